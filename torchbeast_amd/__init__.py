"""torchbeast_amd — an MI355X-native IMPALA actor-learner framework.

A from-scratch reimplementation of the capabilities of
facebookresearch/torchbeast (IMPALA, Espeholt et al. 2018), designed for
AMD Instinct MI355X (gfx950 / CDNA4):

- PyTorch-ROCm is the autograd/driver layer.
- The hot model/RL math (AtariNet conv stack, fused LSTM with done-masking,
  V-trace returns, losses, RMSProp) are hand-written CDNA4 HIP kernels
  (`torchbeast_amd/ops/hip/`), exposed through `torch.autograd.Function`.
- The actor/batching runtime is a native C++ extension (`_tbruntime`) built
  around HIP-pinned rollout ring buffers instead of CPU-side `torch::cat`.
- Multi-GPU training is one process per GPU with `torch.distributed` over
  RCCL/xGMI (flat-gradient all-reduce; see `torchbeast_amd/parallel/`).

Public surface mirrors the reference (capability parity is checked against
the reference layout documented in SURVEY.md):
- `torchbeast_amd.monobeast`  — single-machine trainer (ref: torchbeast/monobeast.py)
- `torchbeast_amd.polybeast_learner` / `polybeast_env` / `polybeast`
  (ref: torchbeast/polybeast*.py)
- `torchbeast_amd.core.vtrace` (ref: torchbeast/core/vtrace.py)
- `libtorchbeast`-equivalent runtime: `torchbeast_amd.runtime`
- `nest`-equivalent: `torchbeast_amd.nest`
"""

__version__ = "0.1.0"
