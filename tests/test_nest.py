"""nest structural ops (ref test strategy: nest/nest_test.py)."""

import pytest

from torchbeast_amd import nest


def test_map():
    n = {"b": (1, 2), "a": [3, {"x": 4}]}
    out = nest.map(lambda v: v * 10, n)
    assert out == {"b": (10, 20), "a": [30, {"x": 40}]}


def test_flatten_sorted_dict_order():
    n = {"b": (1, 2), "a": [3, 4]}
    assert nest.flatten(n) == [3, 4, 1, 2]


def test_pack_as_roundtrip():
    n = {"b": (1, 2), "a": [3, {"x": 4}]}
    flat = nest.flatten(n)
    assert nest.pack_as(n, flat) == n


def test_pack_as_count_mismatch():
    with pytest.raises(ValueError):
        nest.pack_as((1, 2), [1])
    with pytest.raises(ValueError):
        nest.pack_as((1, 2), [1, 2, 3])


def test_map_many2():
    a = (1, {"k": 2})
    b = (10, {"k": 20})
    assert nest.map_many2(lambda x, y: x + y, a, b) == (11, {"k": 22})


def test_map_many_mismatch_raises():
    with pytest.raises(ValueError):
        nest.map_many(lambda leaves: sum(leaves), (1, 2), (1, 2, 3))


def test_front():
    assert nest.front({"z": (5, 6), "a": 7}) == 7
    with pytest.raises(ValueError):
        nest.front(())


def test_leaves_can_be_arbitrary_objects():
    class Blob:
        pass

    blob = Blob()
    n = (blob, [blob])
    flat = nest.flatten(n)
    assert flat == [blob, blob]
    assert nest.pack_as(n, flat) == (blob, [blob])
