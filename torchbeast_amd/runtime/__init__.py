"""Native runtime: queues, dynamic batcher, actor pool, env server.

This package is the `libtorchbeast` equivalent. Python surface (ref:
src/py/__init__.py): BatchingQueue, DynamicBatcher (+Batch), ActorPool,
Server, ClosedBatchingQueue, AsyncError, NestError.

Build with `python setup.py build_ext --inplace` (the extension is built
in-tree so it travels with the repo snapshot).
"""

import torch  # noqa: F401  (loads libc10/libtorch before the extension)

try:
    from torchbeast_amd.runtime._tbruntime import (  # noqa: F401
        ActorPool,
        AsyncError,
        BatchingQueue,
        ClosedBatchingQueue,
        DynamicBatcher,
        NestError,
        Server,
    )

    Batch = DynamicBatcher.Batch
    AVAILABLE = True
except ImportError as _e:  # pragma: no cover - exercised only pre-build
    AVAILABLE = False
    _import_error = _e

    def __getattr__(name):
        raise ImportError(
            f"torchbeast_amd.runtime.{name} requires the _tbruntime extension; "
            "build it with `python setup.py build_ext --inplace`"
        ) from _import_error
