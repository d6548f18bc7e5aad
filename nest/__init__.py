"""Standalone `nest` package (API parity with the reference's nest pip
package, ref: nest/nest/nest_pybind.cc:43-80): map/map_many/map_many2/
flatten/pack_as/front over arbitrarily nested tuples/lists/dicts.

Implementation lives in torchbeast_amd.nest (native C++ via _tbruntime
with a pure-Python fallback); this top-level package makes `import nest`
work exactly as with the reference.
"""

from torchbeast_amd.nest import (  # noqa: F401
    flatten,
    front,
    map,
    map_many,
    map_many2,
    pack_as,
)
