"""MFMA implicit-GEMM conv trunk vs fp32 eager oracle (F.conv2d).

The bf16 kernels (ops/hip/conv_mfma.hip) round their operands to bf16 and
accumulate in fp32; tolerances below are set for that operand rounding
(values are O(1) after /255 normalization; activations |x| < ~3).
"""

import pytest
import torch
import torch.nn.functional as F

pytestmark = pytest.mark.gpu

if torch.cuda.is_available():
    from torchbeast_amd.ops import functional as tbf
    import torchbeast_amd.ops as ops_mod
else:  # pragma: no cover
    pytest.skip("requires ROCm GPU", allow_module_level=True)


def _make_convs(seed=0, device="cuda"):
    torch.manual_seed(seed)
    c1 = torch.nn.Conv2d(4, 32, 8, stride=4).to(device)
    c2 = torch.nn.Conv2d(32, 64, 4, stride=2).to(device)
    c3 = torch.nn.Conv2d(64, 64, 3, stride=1).to(device)
    return c1, c2, c3


def _eager_trunk(frames, c1, c2, c3):
    x = frames.float() / 255.0
    x = F.relu(c1(x))
    x = F.relu(c2(x))
    x = F.relu(c3(x))
    return x.reshape(frames.shape[0], -1)


@pytest.mark.parametrize("N", [1, 7, 32, 500])
def test_trunk_fwd_matches_eager(N):
    c1, c2, c3 = _make_convs()
    torch.manual_seed(N)
    frames = torch.randint(0, 256, (N, 4, 84, 84), dtype=torch.uint8,
                           device="cuda")
    ext = ops_mod.require_ext()
    w1p, w2p, w3p = tbf._pack_trunk_weights(c1.weight, c2.weight, c3.weight)
    (out,) = ext.conv_trunk_fwd(frames, w1p, c1.bias.detach().contiguous(),
                                w2p, c2.bias.detach().contiguous(), w3p,
                                c3.bias.detach().contiguous(), False)
    with torch.no_grad():
        ref = _eager_trunk(frames, c1, c2, c3)
    torch.testing.assert_close(out, ref, rtol=5e-2, atol=3e-2)
    # bf16 rounding must not blow up in aggregate either.
    denom = ref.abs().mean().clamp_min(1e-3)
    assert (out - ref).abs().mean() / denom < 5e-3


@pytest.mark.parametrize("N", [5, 96])
def test_trunk_backward_matches_eager(N):
    c1, c2, c3 = _make_convs(seed=3)
    c1e, c2e, c3e = _make_convs(seed=3)
    torch.manual_seed(N + 100)
    frames = torch.randint(0, 256, (N, 4, 84, 84), dtype=torch.uint8,
                           device="cuda")
    d_out = torch.randn(N, 3136, device="cuda")

    out = tbf._AtariTrunkMfma.apply(
        frames, c1.weight, c1.bias, c2.weight, c2.bias, c3.weight, c3.bias)
    out.backward(d_out)

    ref = _eager_trunk(frames, c1e, c2e, c3e)
    ref.backward(d_out)

    for ours, theirs, name in [
        (c3.weight.grad, c3e.weight.grad, "dw3"),
        (c3.bias.grad, c3e.bias.grad, "db3"),
        (c2.weight.grad, c2e.weight.grad, "dw2"),
        (c2.bias.grad, c2e.bias.grad, "db2"),
        (c1.weight.grad, c1e.weight.grad, "dw1"),
        (c1.bias.grad, c1e.bias.grad, "db1"),
    ]:
        scale = theirs.abs().max().clamp_min(1e-4)
        err = (ours - theirs).abs().max() / scale
        assert err < 3e-2, f"{name}: rel-max err {err:.4f}"


def test_atari_trunk_dispatches_mfma_for_learner_batch():
    c1, c2, c3 = _make_convs(seed=5)
    frames = torch.randint(0, 256, (512, 4, 84, 84), dtype=torch.uint8,
                           device="cuda")
    out = tbf.atari_trunk(frames, c1, c2, c3)
    assert out is not None and out.shape == (512, 3136)
    assert out.requires_grad
    # And grads flow end to end through the custom Function.
    out.sum().backward()
    assert c1.weight.grad is not None and torch.isfinite(c1.weight.grad).all()


def test_trunk_fwd_nograd_large_batch_uses_mfma():
    c1, c2, c3 = _make_convs(seed=6)
    frames = torch.randint(0, 256, (400, 4, 84, 84), dtype=torch.uint8,
                           device="cuda")
    with torch.no_grad():
        out = tbf.atari_trunk(frames, c1, c2, c3)
        ref = _eager_trunk(frames, c1, c2, c3)
    torch.testing.assert_close(out, ref, rtol=5e-2, atol=3e-2)
