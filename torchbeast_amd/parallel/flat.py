"""Flat parameter/gradient layout — the MI355X-native optimizer substrate.

Instead of per-tensor optimizer loops, grad clipping, weight syncs and
all-reduces (the reference does all four per-parameter, e.g. RMSProp at
monobeast.py:387-393, clip at :291, weight copy at polybeast_learner.py:369),
every parameter of the model is re-homed into ONE contiguous buffer:

- gradients accumulate into one flat buffer -> data-parallel training is a
  single RCCL all-reduce over xGMI per step (torchbeast_amd/parallel/ddp.py);
- global-norm clip + RMSProp + LR decay run as one fused HIP kernel over the
  flat buffer (torchbeast_amd.ops.rmsprop_step);
- behavior-model weight sync is one flat device copy (or one RCCL
  broadcast), not a state_dict walk.
"""

import torch

from torchbeast_amd.ops import functional as tbops


def flatten_parameters(module: torch.nn.Module) -> torch.Tensor:
    """Re-home every parameter of `module` into one contiguous flat buffer.

    Returns the flat buffer; module parameters become views into it, so
    state_dict(), forward and autograd all keep working.
    """
    params = [p for p in module.parameters() if p.requires_grad]
    if not params:
        raise ValueError("module has no trainable parameters")
    total = sum(p.numel() for p in params)
    flat = torch.empty(total, device=params[0].device, dtype=params[0].dtype)
    offset = 0
    for p in params:
        n = p.numel()
        flat[offset : offset + n].copy_(p.data.reshape(-1))
        p.data = flat[offset : offset + n].view_as(p.data)
        offset += n
    return flat


def attach_flat_grads(module: torch.nn.Module) -> torch.Tensor:
    """Point every parameter's .grad at a slice of one flat gradient buffer.

    Autograd accumulates in place into existing .grad tensors, so after
    backward() the flat buffer IS the full gradient. Callers must zero it
    with flat.zero_() (never optimizer.zero_grad(set_to_none=True), which
    would drop the views).
    """
    params = [p for p in module.parameters() if p.requires_grad]
    total = sum(p.numel() for p in params)
    flat = torch.zeros(total, device=params[0].device, dtype=params[0].dtype)
    offset = 0
    for p in params:
        n = p.numel()
        p.grad = flat[offset : offset + n].view_as(p.data)
        offset += n
    return flat


class FusedRMSProp:
    """RMSProp over a flat parameter buffer, fused with global-norm clipping
    and the reference's linear LR decay (ref hyperparams:
    monobeast.py:387-398). On GPU the whole step is one HIP kernel pass.

    Matches torch.optim.RMSprop semantics for momentum=0, centered=False.
    """

    def __init__(self, flat_param, flat_grad, lr, alpha=0.99, eps=0.01,
                 clip_norm=None):
        self.param = flat_param
        self.grad = flat_grad
        self.base_lr = lr
        self.alpha = alpha
        self.eps = eps
        self.clip_norm = clip_norm
        self.square_avg = torch.zeros_like(flat_param)
        self.lr_factor = 1.0
        self.steps = 0
        self.last_grad_norm = None
        self.lr_dev = None  # device-resident lr (hipGraph capture mode)

    @property
    def lr(self):
        return self.base_lr * self.lr_factor

    def zero_grad(self):
        self.grad.zero_()

    def enable_device_lr(self):
        """hipGraph mode: the update kernel reads lr from device memory so
        the linear decay keeps applying across graph replays (by-value
        kernel args are frozen at capture)."""
        self.lr_dev = torch.zeros(1, device=self.param.device,
                                  dtype=torch.float32)
        self.push_lr()

    def push_lr(self):
        if self.lr_dev is not None:
            self.lr_dev.fill_(self.lr)

    def step(self):
        self.last_grad_norm = tbops.rmsprop_step(
            self.param,
            self.grad,
            self.square_avg,
            self.lr,
            self.alpha,
            self.eps,
            self.clip_norm,
            self.lr_dev,
        )
        self.steps += 1

    def state_dict(self):
        return {
            "square_avg": self.square_avg,
            "steps": self.steps,
            "lr_factor": self.lr_factor,
            "base_lr": self.base_lr,
            "alpha": self.alpha,
            "eps": self.eps,
            "clip_norm": self.clip_norm,
        }

    def load_state_dict(self, state):
        self.square_avg.copy_(state["square_avg"])
        self.steps = state["steps"]
        self.lr_factor = state["lr_factor"]
        self.base_lr = state["base_lr"]
        self.alpha = state["alpha"]
        self.eps = state["eps"]
        self.clip_norm = state["clip_norm"]


class LinearLR:
    """Linear decay to zero over total_steps env steps (ref:
    monobeast.py:395-398), driving a FusedRMSProp's lr_factor."""

    def __init__(self, optimizer: FusedRMSProp, steps_per_update, total_steps):
        self.optimizer = optimizer
        self.steps_per_update = steps_per_update
        self.total_steps = total_steps
        self.updates = 0

    def step(self):
        self.updates += 1
        done = min(self.updates * self.steps_per_update, self.total_steps)
        self.optimizer.lr_factor = 1.0 - done / self.total_steps

    def state_dict(self):
        return {"updates": self.updates}

    def load_state_dict(self, state):
        self.updates = state["updates"]
