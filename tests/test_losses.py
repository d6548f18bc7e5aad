"""Loss values and analytic gradients (ref test strategy:
tests/polybeast_loss_functions_test.py). The analytic-gradient checks here
are the same formulas the fused HIP loss kernel implements, so they double
as its CPU oracle."""

import numpy as np
import torch

from torchbeast_amd.core import losses
from torchbeast_amd.ops import functional as tbops


def _softmax(x):
    e = np.exp(x - x.max(-1, keepdims=True))
    return e / e.sum(-1, keepdims=True)


def test_baseline_loss_value_and_grad():
    adv = torch.randn(6, 3, requires_grad=True)
    loss = losses.compute_baseline_loss(adv)
    expected = 0.5 * (adv.detach().numpy() ** 2).sum()
    np.testing.assert_allclose(loss.item(), expected, rtol=1e-5)

    loss.backward()
    np.testing.assert_allclose(adv.grad.numpy(), adv.detach().numpy(), rtol=1e-5)


def test_entropy_loss_value_and_grad():
    logits = torch.randn(5, 2, 7, requires_grad=True)
    loss = losses.compute_entropy_loss(logits)

    z = logits.detach().numpy()
    p = _softmax(z)
    logp = z - np.log(np.exp(z - z.max(-1, keepdims=True)).sum(-1, keepdims=True)) - z.max(-1, keepdims=True)
    expected = (p * logp).sum()
    np.testing.assert_allclose(loss.item(), expected, rtol=1e-4)

    loss.backward()
    # d/dz sum(p log p) = p * (log p - sum_a p_a log p_a)  (per row)
    neg_ent_row = (p * logp).sum(-1, keepdims=True)
    expected_grad = p * (logp - neg_ent_row)
    np.testing.assert_allclose(logits.grad.numpy(), expected_grad, rtol=1e-4, atol=1e-6)


def test_policy_gradient_loss_value_and_grad():
    T, B, A = 4, 3, 6
    logits = torch.randn(T, B, A, requires_grad=True)
    actions = torch.randint(0, A, (T, B))
    advantages = torch.randn(T, B, requires_grad=True)

    loss = losses.compute_policy_gradient_loss(logits, actions, advantages)

    z = logits.detach().numpy()
    p = _softmax(z)
    a = actions.numpy()
    adv = advantages.detach().numpy()
    ce = -np.log(np.take_along_axis(p, a[..., None], axis=-1).squeeze(-1))
    np.testing.assert_allclose(loss.item(), (ce * adv).sum(), rtol=1e-4)

    loss.backward()
    # d/dz ce(z, a)*adv = adv * (softmax(z) - onehot(a))
    onehot = np.eye(A)[a]
    expected_grad = adv[..., None] * (p - onehot)
    np.testing.assert_allclose(logits.grad.numpy(), expected_grad, rtol=1e-4, atol=1e-6)
    # Advantages are detached inside the loss: no gradient may flow to them.
    assert advantages.grad is None


def test_fused_impala_loss_eager_path_matches_components():
    T, B, A = 5, 4, 9
    logits = torch.randn(T, B, A, requires_grad=True)
    baseline = torch.randn(T, B, requires_grad=True)
    actions = torch.randint(0, A, (T, B))
    pg_adv = torch.randn(T, B)
    vs = torch.randn(T, B)

    pg, bl, ent = tbops.fused_impala_loss(logits, baseline, actions, pg_adv, vs)
    torch.testing.assert_close(
        pg, losses.compute_policy_gradient_loss(logits, actions, pg_adv)
    )
    torch.testing.assert_close(bl, losses.compute_baseline_loss(vs - baseline))
    torch.testing.assert_close(ent, losses.compute_entropy_loss(logits))

    total = pg + 0.5 * bl + 0.0006 * ent
    total.backward()
    assert logits.grad is not None and baseline.grad is not None
    # Baseline-loss gradient is (baseline - vs) * cost.
    torch.testing.assert_close(baseline.grad, 0.5 * (baseline.detach() - vs))
