"""V-trace: off-policy corrected value targets (IMPALA, arXiv:1802.01561 §4.1).

API-compatible with the reference implementation (ref:
torchbeast/core/vtrace.py:50-139) but written MI355X-first: on CUDA/ROCm
tensors, the whole pipeline — action log-probs for behavior+target policies,
importance-weight clipping, the reverse-time scan over T, and the
policy-gradient advantages — runs as a single fused HIP kernel
(`torchbeast_amd.ops.vtrace_fused`), one workgroup per batch column, instead
of a T-step Python loop. This module keeps a plain PyTorch implementation as
the CPU path and the numerics oracle.

All returned tensors carry no gradient (targets are constants w.r.t. the
learner); gradients flow through the losses' use of the *learner* logits and
baseline instead.
"""

import collections

import torch
import torch.nn.functional as F

VTraceFromLogitsReturns = collections.namedtuple(
    "VTraceFromLogitsReturns",
    [
        "vs",
        "pg_advantages",
        "log_rhos",
        "behavior_action_log_probs",
        "target_action_log_probs",
    ],
)

VTraceReturns = collections.namedtuple("VTraceReturns", "vs pg_advantages")


def action_log_probs(policy_logits, actions):
    """log pi(a_t | x_t) for the given actions, shape = actions.shape."""
    log_pi = F.log_softmax(policy_logits, dim=-1)
    return log_pi.gather(-1, actions.unsqueeze(-1)).squeeze(-1)


def from_logits(
    behavior_policy_logits,
    target_policy_logits,
    actions,
    discounts,
    rewards,
    values,
    bootstrap_value,
    clip_rho_threshold=1.0,
    clip_pg_rho_threshold=1.0,
):
    """V-trace for softmax policies, from raw logits.

    Inputs are time-major: logits [T, B, A], everything else [T, B]
    (bootstrap_value [B]).
    """
    if behavior_policy_logits.is_cuda:
        from torchbeast_amd import ops

        if ops.hip_available():
            return ops.vtrace_from_logits(
                behavior_policy_logits,
                target_policy_logits,
                actions,
                discounts,
                rewards,
                values,
                bootstrap_value,
                clip_rho_threshold,
                clip_pg_rho_threshold,
            )

    target_lp = action_log_probs(target_policy_logits, actions)
    behavior_lp = action_log_probs(behavior_policy_logits, actions)
    log_rhos = target_lp - behavior_lp
    core = from_importance_weights(
        log_rhos=log_rhos,
        discounts=discounts,
        rewards=rewards,
        values=values,
        bootstrap_value=bootstrap_value,
        clip_rho_threshold=clip_rho_threshold,
        clip_pg_rho_threshold=clip_pg_rho_threshold,
    )
    return VTraceFromLogitsReturns(
        vs=core.vs,
        pg_advantages=core.pg_advantages,
        log_rhos=log_rhos,
        behavior_action_log_probs=behavior_lp,
        target_action_log_probs=target_lp,
    )


@torch.no_grad()
def from_importance_weights(
    log_rhos,
    discounts,
    rewards,
    values,
    bootstrap_value,
    clip_rho_threshold=1.0,
    clip_pg_rho_threshold=1.0,
):
    """V-trace from log importance weights log_rhos [T, B].

    vs_t = V(x_t) + sum_{k>=t} gamma^{k-t} (prod_{i<k} c_i) delta_k V,
    computed as the reverse recurrence
        acc_t = delta_t + discount_t * c_t * acc_{t+1}.
    """
    rhos = torch.exp(log_rhos)
    if clip_rho_threshold is not None:
        clipped_rhos = rhos.clamp(max=clip_rho_threshold)
    else:
        clipped_rhos = rhos
    cs = rhos.clamp(max=1.0)

    next_values = torch.cat([values[1:], bootstrap_value.unsqueeze(0)], dim=0)
    deltas = clipped_rhos * (rewards + discounts * next_values - values)

    vs_minus_v = torch.empty_like(deltas)
    acc = torch.zeros_like(bootstrap_value)
    for t in range(deltas.shape[0] - 1, -1, -1):
        acc = deltas[t] + discounts[t] * cs[t] * acc
        vs_minus_v[t] = acc

    vs = vs_minus_v + values

    next_vs = torch.cat([vs[1:], bootstrap_value.unsqueeze(0)], dim=0)
    if clip_pg_rho_threshold is not None:
        clipped_pg_rhos = rhos.clamp(max=clip_pg_rho_threshold)
    else:
        clipped_pg_rhos = rhos
    pg_advantages = clipped_pg_rhos * (rewards + discounts * next_vs - values)

    return VTraceReturns(vs=vs, pg_advantages=pg_advantages)
