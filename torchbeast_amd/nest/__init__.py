"""nest — structural operations over nested tuples/lists/dicts of leaves.

API parity with the reference's standalone `nest` package
(ref: nest/nest/nest_pybind.cc:43-80): map, map_many, map_many2, flatten,
pack_as, front. Dict keys are visited in sorted order (the reference's C++
`std::map` ordering), so flatten/pack_as round-trips are stable across the
Python and C++ implementations.

The C++ implementation lives in the `_tbruntime` extension (same functions
over py::object trees, plus the TensorNest used inside the runtime); when it
is importable its functions replace the pure-Python ones below.
"""


def _is_leaf(obj):
    return not isinstance(obj, (tuple, list, dict))


def map(f, nest):  # noqa: A001 - name fixed by the reference API
    if isinstance(nest, dict):
        return {k: map(f, nest[k]) for k in nest}
    if isinstance(nest, tuple):
        return tuple(map(f, v) for v in nest)
    if isinstance(nest, list):
        return [map(f, v) for v in nest]
    return f(nest)


def map_many(f, *nests):
    if not nests:
        raise ValueError("map_many requires at least one nest")
    first = nests[0]
    if isinstance(first, dict):
        keys = list(first)
        for n in nests[1:]:
            if not isinstance(n, dict) or set(n) != set(keys):
                raise ValueError("nests don't match")
        return {k: map_many(f, *[n[k] for n in nests]) for k in keys}
    if isinstance(first, (tuple, list)):
        length = len(first)
        for n in nests[1:]:
            if not isinstance(n, type(first)) or len(n) != length:
                raise ValueError("nests don't match")
        mapped = [map_many(f, *[n[i] for n in nests]) for i in range(length)]
        return tuple(mapped) if isinstance(first, tuple) else mapped
    for n in nests[1:]:
        if not _is_leaf(n):
            raise ValueError("nests don't match")
    return f(list(nests))


def map_many2(f, nest1, nest2):
    return map_many(lambda pair: f(pair[0], pair[1]), nest1, nest2)


def flatten(nest):
    out = []

    def visit(obj):
        if isinstance(obj, dict):
            for k in sorted(obj):
                visit(obj[k])
        elif isinstance(obj, (tuple, list)):
            for v in obj:
                visit(v)
        else:
            out.append(obj)

    visit(nest)
    return out


def pack_as(nest, flat_sequence):
    flat = list(flat_sequence)
    pos = 0

    def build(obj):
        nonlocal pos
        if isinstance(obj, dict):
            # Fill in sorted-key order, then restore the original key order.
            filled = {}
            for k in sorted(obj):
                filled[k] = build(obj[k])
            return {k: filled[k] for k in obj}
        if isinstance(obj, (tuple, list)):
            built = [build(v) for v in obj]
            return tuple(built) if isinstance(obj, tuple) else built
        if pos >= len(flat):
            raise ValueError("Too few elements to pack")
        leaf = flat[pos]
        pos += 1
        return leaf

    result = build(nest)
    if pos != len(flat):
        raise ValueError("Too many elements to pack")
    return result


def front(nest):
    f = flatten(nest)
    if not f:
        raise ValueError("front() of empty nest")
    return f[0]


try:  # Prefer the native implementation when the runtime extension is built.
    from torchbeast_amd.runtime import _tbruntime as _native

    map = _native.map  # noqa: A001,F811
    map_many = _native.map_many  # noqa: F811
    map_many2 = _native.map_many2  # noqa: F811
    flatten = _native.flatten  # noqa: F811
    pack_as = _native.pack_as  # noqa: F811
    front = _native.front  # noqa: F811
except ImportError:
    pass
