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


def _make_convs(seed=0, device="cuda", in_ch=4):
    torch.manual_seed(seed)
    c1 = torch.nn.Conv2d(in_ch, 32, 8, stride=4).to(device)
    c2 = torch.nn.Conv2d(32, 64, 4, stride=2).to(device)
    c3 = torch.nn.Conv2d(64, 64, 3, stride=1).to(device)
    return c1, c2, c3


def _eager_trunk(frames, c1, c2, c3):
    x = frames.float() / 255.0
    x = F.relu(c1(x))
    x = F.relu(c2(x))
    x = F.relu(c3(x))
    return x.reshape(frames.shape[0], -1)


@pytest.mark.parametrize("N,shape", [(1, (4, 84, 84)), (7, (4, 84, 84)),
                                     (32, (4, 84, 84)), (500, (4, 84, 84)),
                                     (3, (3, 210, 160)), (40, (3, 210, 160))])
def test_trunk_fwd_matches_eager(N, shape):
    c1, c2, c3 = _make_convs(in_ch=shape[0])
    torch.manual_seed(N)
    frames = torch.randint(0, 256, (N, *shape), dtype=torch.uint8,
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


def _bf(x):
    return x.to(torch.bfloat16).float()


def _semi_oracle_grads(frames, d_out, c1, c2, c3, a1, a2, out3):
    """Eager backward that rounds operands to bf16 exactly where the HIP
    pipeline does (MFMA rounds inputs, accumulates fp32) and reuses the
    KERNEL's saved activations for the ReLU masks — so any remaining
    mismatch is an indexing bug, not mixed-precision divergence."""
    from torch.nn.grad import conv2d_input, conv2d_weight

    a1f = a1.permute(0, 3, 1, 2).float()  # NHWC bf16 -> NCHW fp32
    a2f = a2.permute(0, 3, 1, 2).float()
    N = frames.shape[0]
    oh3 = a2.shape[1] - 2
    ow3 = a2.shape[2] - 2
    d3 = (d_out * (out3 > 0)).view(N, 64, oh3, ow3)
    d3 = _bf(d3)
    dw3 = conv2d_weight(_bf(a2f), c3.weight.shape, d3)
    db3 = d3.sum((0, 2, 3))
    d2 = conv2d_input(a2f.shape, _bf(c3.weight), d3) * (a2f > 0)
    d2 = _bf(d2)
    dw2 = conv2d_weight(_bf(a1f), c2.weight.shape, d2, stride=2)
    db2 = d2.sum((0, 2, 3))
    d1 = conv2d_input(a1f.shape, _bf(c2.weight), d2, stride=2) * (a1f > 0)
    d1 = _bf(d1)
    x = _bf(frames.float() / 255.0)
    dw1 = conv2d_weight(x, c1.weight.shape, d1, stride=4)
    db1 = d1.sum((0, 2, 3))
    return dw1, db1, dw2, db2, dw3, db3


@pytest.mark.parametrize("N,shape", [(5, (4, 84, 84)), (96, (4, 84, 84)),
                                     (6, (3, 210, 160))])
def test_trunk_backward_matches_eager(N, shape):
    c1, c2, c3 = _make_convs(seed=3, in_ch=shape[0])
    torch.manual_seed(N + 100)
    frames = torch.randint(0, 256, (N, *shape), dtype=torch.uint8,
                           device="cuda")
    with torch.no_grad():
        nfeat = _eager_trunk(frames[:1], c1, c2, c3).shape[1]
    d_out = torch.randn(N, nfeat, device="cuda")

    # Kernel path (grab the stash for the semi-oracle).
    ext = ops_mod.require_ext()
    w1p, w2p, w3p = tbf._pack_trunk_weights(c1.weight, c2.weight, c3.weight)
    out3, a1, a2 = ext.conv_trunk_fwd(
        frames, w1p, c1.bias.detach().contiguous(), w2p,
        c2.bias.detach().contiguous(), w3p, c3.bias.detach().contiguous(),
        True)
    out = tbf._AtariTrunkMfma.apply(
        frames, c1.weight, c1.bias, c2.weight, c2.bias, c3.weight, c3.bias)
    out.backward(d_out)

    oracle = _semi_oracle_grads(frames, d_out, c1, c2, c3, a1, a2, out3)
    names = ["dw1", "db1", "dw2", "db2", "dw3", "db3"]
    got = [c1.weight.grad, c1.bias.grad, c2.weight.grad, c2.bias.grad,
           c3.weight.grad, c3.bias.grad]
    for ours, theirs, name in zip(got, oracle, names):
        scale = theirs.abs().max().clamp_min(1e-4)
        err = (ours - theirs).abs().max() / scale
        # Operand bf16 rounding differs only in summation order now.
        assert err < 1e-2, f"{name}: rel-max err {err:.4f}"

    # Against the true fp32 oracle the gradient direction must still agree.
    c1e, c2e, c3e = _make_convs(seed=3, in_ch=shape[0])
    ref = _eager_trunk(frames, c1e, c2e, c3e)
    ref.backward(d_out)
    for ours, theirs, name in [
        (c3.weight.grad, c3e.weight.grad, "dw3"),
        (c2.weight.grad, c2e.weight.grad, "dw2"),
        (c1.weight.grad, c1e.weight.grad, "dw1"),
    ]:
        cos = torch.nn.functional.cosine_similarity(
            ours.flatten(), theirs.flatten(), dim=0)
        assert cos > 0.995, f"{name}: cosine {cos:.5f}"


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
