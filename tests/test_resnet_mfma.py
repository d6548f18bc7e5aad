"""Deep-ResNet MFMA 3x3 conv kernels vs eager oracles (ops/hip/conv_mfma.hip
resnet_conv / resnet_conv_wgrad; model wiring in models/resnet.py).

Forward compares against fp32 F.conv2d with bf16-rounded operands;
backward uses a precision-faithful semi-oracle like test_conv_mfma.py.
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

GEOMS = [(16, 42, 16), (16, 42, 32), (32, 21, 32), (32, 11, 32)]


def _bf(x):
    return x.to(torch.bfloat16).float()


@pytest.mark.parametrize("ci,hw,co", GEOMS)
@pytest.mark.parametrize("n", [1, 13])
def test_resnet_conv_fwd(ci, hw, co, n):
    torch.manual_seed(ci * hw + co + n)
    conv = torch.nn.Conv2d(ci, co, 3, padding=1).cuda()
    x = (torch.randn(n, ci, hw, hw, device="cuda") * 0.5).to(torch.bfloat16)
    xcl = x.contiguous(memory_format=torch.channels_last)
    with torch.no_grad():
        out = tbf.resnet_conv3x3(conv, xcl)
        ref = F.conv2d(_bf(x), _bf(conv.weight), conv.bias, padding=1)
    assert out.shape == ref.shape
    torch.testing.assert_close(out.float(), ref, rtol=5e-2, atol=2e-2)


def test_resnet_first_conv_fwd_and_wgrad():
    # Obs channels zero-padded to 8; backward is wgrad-only.
    torch.manual_seed(11)
    conv = torch.nn.Conv2d(4, 16, 3, padding=1).cuda()
    x = torch.rand(7, 4, 84, 84, device="cuda")
    out = tbf.resnet_first_conv(conv, x)
    with torch.no_grad():
        ref = F.conv2d(_bf(x), _bf(conv.weight), conv.bias, padding=1)
    torch.testing.assert_close(out.float(), ref, rtol=5e-2, atol=2e-2)

    dy = torch.randn_like(out)
    out.backward(dy)
    from torch.nn.grad import conv2d_weight
    dyb = _bf(dy.float())
    dw_ref = conv2d_weight(_bf(x), conv.weight.shape, dyb, padding=1)
    db_ref = dyb.sum((0, 2, 3))
    for got, ref_g, name in [(conv.weight.grad, dw_ref, "dw"),
                             (conv.bias.grad, db_ref, "db")]:
        scale = ref_g.abs().max().clamp_min(1e-4)
        err = (got.float() - ref_g).abs().max() / scale
        assert err < 2e-2, f"{name}: rel-max err {err:.4f}"


def test_resnet_dgrad_geometry_42():
    # The section-2 feature conv's dgrad runs the 32->16 @42 geometry.
    torch.manual_seed(7)
    conv = torch.nn.Conv2d(16, 32, 3, padding=1).cuda()
    x = (torch.randn(5, 16, 42, 42, device="cuda") * 0.5).to(
        torch.bfloat16).contiguous(memory_format=torch.channels_last)
    x.requires_grad_(True)
    out = tbf.resnet_conv3x3(conv, x)
    dy = torch.randn_like(out)
    out.backward(dy)
    from torch.nn.grad import conv2d_input
    ref = conv2d_input(x.shape, _bf(conv.weight), _bf(dy).float(), padding=1)
    torch.testing.assert_close(x.grad.float(), _bf(ref.to(torch.bfloat16)),
                               rtol=5e-2, atol=2e-2)


@pytest.mark.parametrize("ci,hw,co", GEOMS)
def test_resnet_conv_backward(ci, hw, co):
    from torch.nn.grad import conv2d_input, conv2d_weight

    torch.manual_seed(ci + hw + co)
    conv = torch.nn.Conv2d(ci, co, 3, padding=1).cuda()
    n = 9
    x = (torch.randn(n, ci, hw, hw, device="cuda") * 0.5).to(
        torch.bfloat16).contiguous(memory_format=torch.channels_last)
    x.requires_grad_(True)
    out = tbf.resnet_conv3x3(conv, x)
    dy = torch.randn_like(out)
    out.backward(dy)

    dyb = _bf(dy.float())
    xb = _bf(x.detach().float())
    dw_ref = conv2d_weight(xb, conv.weight.shape, dyb, padding=1)
    db_ref = dyb.sum((0, 2, 3))
    dx_ref = conv2d_input(x.shape, _bf(conv.weight), dyb, padding=1)

    for got, ref, name in [(conv.weight.grad, dw_ref, "dw"),
                           (conv.bias.grad, db_ref, "db"),
                           (x.grad.float(), dx_ref, "dx")]:
        scale = ref.abs().max().clamp_min(1e-4)
        err = (got.float() - ref).abs().max() / scale
        assert err < 2e-2, f"{name}: rel-max err {err:.4f}"


def test_resnet_model_trunk_matches_eager():
    from torchbeast_amd.models.resnet import ResNet

    torch.manual_seed(0)
    net = ResNet((4, 84, 84), 6).cuda()
    x = torch.randn(6, 4, 84, 84, device="cuda")
    fused = net._features_mfma(x)
    with torch.no_grad():
        ref = net.feat_extract(x)
    assert fused.shape == ref.shape
    cos = F.cosine_similarity(fused.flatten(), ref.flatten(), dim=0)
    assert cos > 0.999, f"trunk cosine {cos:.5f}"
    # Aggregate error small relative to activation scale.
    denom = ref.abs().mean().clamp_min(1e-3)
    assert (fused - ref).abs().mean() / denom < 5e-2


def test_resnet_model_end_to_end_grads():
    from torchbeast_amd.models.resnet import ResNet

    torch.manual_seed(1)
    net = ResNet((4, 84, 84), 6, use_lstm=False).cuda()
    T, B = 4, 8
    inputs = dict(
        frame=torch.randint(0, 256, (T, B, 4, 84, 84), dtype=torch.uint8,
                            device="cuda"),
        reward=torch.randn(T, B, device="cuda"),
        done=torch.zeros(T, B, dtype=torch.bool, device="cuda"),
    )
    (action, logits, baseline), _ = net(inputs, ())
    (logits.float().square().mean() + baseline.float().square().mean()
     ).backward()
    for name, p in net.named_parameters():
        assert p.grad is not None, name
        assert torch.isfinite(p.grad).all(), name

    # Grad direction agrees with the pure-eager path.
    import os
    net2 = ResNet((4, 84, 84), 6).cuda()
    net2.load_state_dict(net.state_dict())
    os.environ["TBAMD_RESNET"] = "aten"
    try:
        torch.manual_seed(123)
        (_, l2, b2), _ = net2(inputs, ())
        (l2.float().square().mean() + b2.float().square().mean()).backward()
    finally:
        del os.environ["TBAMD_RESNET"]
    for (n1, p1), (_, p2) in zip(net.named_parameters(),
                                 net2.named_parameters()):
        if "conv" not in n1 and "feat" not in n1:
            continue
        cos = F.cosine_similarity(p1.grad.flatten(), p2.grad.flatten(), dim=0)
        assert cos > 0.98, f"{n1}: grad cosine {cos:.4f}"
