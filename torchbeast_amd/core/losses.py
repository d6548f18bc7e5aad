"""IMPALA losses (ref: torchbeast/monobeast.py:107-125 ==
torchbeast/polybeast_learner.py:113-131).

Three scalar losses over [T, B] rollout batches:
- baseline:   0.5 * sum((vs - baseline)^2)
- entropy:    sum(softmax(logits) * log_softmax(logits))   (note: +p*log p,
              i.e. the *negative* entropy, weighted positively in the total)
- policy:     sum(ce(logits, actions) * advantages.detach())

On GPU with the HIP extension loaded, the learner uses the fused kernel in
`torchbeast_amd.ops.fused_impala_loss` which computes all three losses AND
their gradients w.r.t. (logits, baseline) in a single pass; this module is
the eager/CPU reference used for tests and for CPU training.
"""

import torch
import torch.nn.functional as F


def compute_baseline_loss(advantages):
    return 0.5 * torch.sum(advantages**2)


def compute_entropy_loss(logits):
    """Return the *negative* entropy, summed over the batch."""
    policy = F.softmax(logits, dim=-1)
    log_policy = F.log_softmax(logits, dim=-1)
    return torch.sum(policy * log_policy)


def compute_policy_gradient_loss(logits, actions, advantages):
    cross_entropy = F.nll_loss(
        F.log_softmax(torch.flatten(logits, 0, 1), dim=-1),
        target=torch.flatten(actions, 0, 1),
        reduction="none",
    )
    cross_entropy = cross_entropy.view_as(advantages)
    return torch.sum(cross_entropy * advantages.detach())
