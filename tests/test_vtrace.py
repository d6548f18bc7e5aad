"""V-trace numerics against an independent NumPy transcription of the paper
formula (ref test strategy: tests/vtrace_test.py)."""

import numpy as np
import pytest
import torch

from torchbeast_amd.core import vtrace


def ground_truth_vtrace(
    log_rhos,
    discounts,
    rewards,
    values,
    bootstrap_value,
    clip_rho_threshold=1.0,
    clip_pg_rho_threshold=1.0,
):
    """O(T^2) literal transcription of arXiv:1802.01561 eq. (1)."""
    rhos = np.exp(log_rhos)
    clipped_rhos = np.minimum(rhos, clip_rho_threshold)
    cs = np.minimum(rhos, 1.0)
    T = rewards.shape[0]

    values_t_plus_1 = np.concatenate([values[1:], bootstrap_value[None]], axis=0)
    deltas = clipped_rhos * (rewards + discounts * values_t_plus_1 - values)

    vs = []
    for s in range(T):
        v_s = values[s].astype(np.float64).copy()
        for t in range(s, T):
            coeff = np.prod(discounts[s:t] * cs[s:t], axis=0) if t > s else 1.0
            v_s = v_s + coeff * deltas[t]
        vs.append(v_s)
    vs = np.stack(vs)

    vs_t_plus_1 = np.concatenate([vs[1:], bootstrap_value[None]], axis=0)
    pg_rhos = np.minimum(rhos, clip_pg_rho_threshold)
    pg_advantages = pg_rhos * (rewards + discounts * vs_t_plus_1 - values)
    return vs, pg_advantages


def _random_inputs(T, B, seed=0):
    rng = np.random.RandomState(seed)
    return dict(
        log_rhos=(rng.uniform(-1.5, 1.5, (T, B))).astype(np.float32),
        discounts=(rng.uniform(0.0, 1.0, (T, B)) > 0.1).astype(np.float32) * 0.99,
        rewards=rng.randn(T, B).astype(np.float32),
        values=rng.randn(T, B).astype(np.float32),
        bootstrap_value=rng.randn(B).astype(np.float32),
    )


@pytest.mark.parametrize("T,B", [(5, 2), (80, 8), (1, 1)])
def test_from_importance_weights_matches_numpy(T, B):
    inp = _random_inputs(T, B)
    expected_vs, expected_pg = ground_truth_vtrace(**inp)

    out = vtrace.from_importance_weights(
        **{k: torch.from_numpy(v) for k, v in inp.items()}
    )
    np.testing.assert_allclose(out.vs.numpy(), expected_vs, rtol=1e-4, atol=1e-4)
    np.testing.assert_allclose(
        out.pg_advantages.numpy(), expected_pg, rtol=1e-4, atol=1e-4
    )


@pytest.mark.parametrize("clip_rho,clip_pg", [(0.5, 0.3), (3.0, 2.0)])
def test_clip_thresholds(clip_rho, clip_pg):
    inp = _random_inputs(7, 3, seed=1)
    expected_vs, expected_pg = ground_truth_vtrace(
        **inp, clip_rho_threshold=clip_rho, clip_pg_rho_threshold=clip_pg
    )
    out = vtrace.from_importance_weights(
        **{k: torch.from_numpy(v) for k, v in inp.items()},
        clip_rho_threshold=clip_rho,
        clip_pg_rho_threshold=clip_pg,
    )
    np.testing.assert_allclose(out.vs.numpy(), expected_vs, rtol=1e-4, atol=1e-4)
    np.testing.assert_allclose(
        out.pg_advantages.numpy(), expected_pg, rtol=1e-4, atol=1e-4
    )


def test_from_logits_consistent_with_importance_weights():
    T, B, A = 6, 4, 5
    rng = np.random.RandomState(2)
    behavior_logits = torch.from_numpy(rng.randn(T, B, A).astype(np.float32))
    target_logits = torch.from_numpy(rng.randn(T, B, A).astype(np.float32))
    actions = torch.from_numpy(rng.randint(0, A, (T, B)))
    inp = _random_inputs(T, B, seed=3)

    out = vtrace.from_logits(
        behavior_logits,
        target_logits,
        actions,
        torch.from_numpy(inp["discounts"]),
        torch.from_numpy(inp["rewards"]),
        torch.from_numpy(inp["values"]),
        torch.from_numpy(inp["bootstrap_value"]),
    )

    # Cross-check: same result through the importance-weights API.
    log_rhos = out.target_action_log_probs - out.behavior_action_log_probs
    torch.testing.assert_close(out.log_rhos, log_rhos)
    core = vtrace.from_importance_weights(
        log_rhos,
        torch.from_numpy(inp["discounts"]),
        torch.from_numpy(inp["rewards"]),
        torch.from_numpy(inp["values"]),
        torch.from_numpy(inp["bootstrap_value"]),
    )
    torch.testing.assert_close(out.vs, core.vs)
    torch.testing.assert_close(out.pg_advantages, core.pg_advantages)


def test_action_log_probs_matches_log_softmax():
    T, B, A = 4, 3, 7
    logits = torch.randn(T, B, A)
    actions = torch.randint(0, A, (T, B))
    lp = vtrace.action_log_probs(logits, actions)
    ref = torch.log_softmax(logits, dim=-1)
    expected = torch.stack(
        [
            torch.stack([ref[t, b, actions[t, b]] for b in range(B)])
            for t in range(T)
        ]
    )
    torch.testing.assert_close(lp, expected)


def test_no_gradients_flow_through_vtrace():
    inp = _random_inputs(5, 2)
    values = torch.from_numpy(inp["values"]).requires_grad_()
    out = vtrace.from_importance_weights(
        torch.from_numpy(inp["log_rhos"]),
        torch.from_numpy(inp["discounts"]),
        torch.from_numpy(inp["rewards"]),
        values,
        torch.from_numpy(inp["bootstrap_value"]),
    )
    assert not out.vs.requires_grad
    assert not out.pg_advantages.requires_grad
