"""Dispatch layer for the hand-written CDNA4 HIP kernels.

The compiled extension (`torchbeast_amd.ops._tbops`, built from
`torchbeast_amd/ops/hip/*.hip` for gfx950 — see setup.py and
`__graft_entry__.build`) provides the device implementations. This module
wraps them in `torch.autograd.Function`s and exposes:

- `vtrace_from_logits(...)`         — fused V-trace (log-probs + clip + scan)
- `fused_impala_loss(...)`          — all three losses + grads in one kernel
- `lstm_unroll(...)`                — done-masked multi-layer LSTM over T
- `rmsprop_step(...)`               — fused global-norm clip + RMSProp
- `policy_sample(...)`              — softmax + multinomial / argmax
- `atari_trunk(...)`                — fused u8→f32 conv stack (AtariNet)

Policy on missing extension: on a machine WITH a GPU the ops refuse to fall
back silently — a missing/unbuilt extension raises ImportError so a broken
build can't masquerade as a working HIP path (set TBAMD_ALLOW_EAGER=1 to
override for debugging). On CPU-only machines the eager reference
implementations are used, and are also the numerics oracles for the GPU
tests (tests/test_ops_gpu.py).
"""

import os

import torch

_tbops = None
_tbops_error = None


def _raise_if_gpu():
    if torch.cuda.is_available() and not os.environ.get("TBAMD_ALLOW_EAGER"):
        raise ImportError(
            "torchbeast_amd HIP extension (_tbops) failed to import but a GPU "
            "is present; build it with `python setup.py build_ext --inplace` "
            "(PYTORCH_ROCM_ARCH=gfx950). Set TBAMD_ALLOW_EAGER=1 to run on "
            f"stock PyTorch ops anyway. Original error: {_tbops_error!r}"
        ) from _tbops_error


def _load():
    global _tbops, _tbops_error
    if _tbops is not None:
        return _tbops
    if _tbops_error is not None:
        _raise_if_gpu()
        return None
    try:
        # importlib, NOT `from torchbeast_amd.ops import _tbops`: this
        # module's own `_tbops = None` global would shadow the submodule.
        import importlib

        _tbops = importlib.import_module("torchbeast_amd.ops._tbops")
    except Exception as e:  # ImportError, OSError (dlopen), RuntimeError (HIP)
        _tbops_error = e
        _raise_if_gpu()
    return _tbops


def hip_available():
    """True iff the gfx950 extension is importable (does not need a GPU)."""
    try:
        return _load() is not None
    except ImportError:
        raise


def require_ext():
    ext = _load()
    if ext is None:
        raise ImportError(
            "torchbeast_amd._tbops is required here but not built "
            f"(original error: {_tbops_error!r})"
        ) from _tbops_error
    return ext


# Fused-op wrappers are defined in submodules to keep this init cheap.
from torchbeast_amd.ops.functional import (  # noqa: E402,F401
    fused_impala_loss,
    lstm_unroll,
    policy_sample,
    rmsprop_step,
    vtrace_from_logits,
)
