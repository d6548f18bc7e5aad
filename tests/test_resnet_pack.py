"""CPU checks of the deep-ResNet MFMA operand layouts
(ops/functional.py _pack_resnet_weight and the wgrad unpack): the GPU
kernels assume a (ky,kx,c)-major K order zero-padded to 32; these tests
pin that contract without needing a GPU."""

import torch
import torch.nn.functional as F

from torchbeast_amd.ops.functional import _pack_resnet_weight


def test_pack_layout_roundtrip():
    torch.manual_seed(0)
    w = torch.randn(32, 16, 3, 3)
    p = _pack_resnet_weight(w)
    k = 3 * 3 * 16
    kp = (k + 31) // 32 * 32
    assert p.shape == (32, kp) and p.dtype == torch.bfloat16
    assert (p[:, k:] == 0).all()
    back = p[:, :k].view(32, 3, 3, 16).permute(0, 3, 1, 2)
    torch.testing.assert_close(back.float(), w.to(torch.bfloat16).float())


def test_pack_rotated_is_dgrad_operand():
    # dgrad = stride-1 conv of the padded dY with flipped taps, channels
    # swapped: W_rot[ci][ky][kx][co] = W[co][ci][2-ky][2-kx].
    torch.manual_seed(1)
    w = torch.randn(32, 16, 3, 3)
    p = _pack_resnet_weight(w, rotate=True)
    k = 3 * 3 * 32
    assert p.shape[0] == 16 and (p[:, k:] == 0).all()
    got = p[:, :k].view(16, 3, 3, 32)
    for ky in (0, 2):
        for kx in (0, 1):
            torch.testing.assert_close(
                got[:, ky, kx, :].float(),
                w[:, :, 2 - ky, 2 - kx].t().to(torch.bfloat16).float())


def test_packed_gemm_matches_conv2d():
    # Emulate the kernel's implicit GEMM with the packed operand on CPU:
    # NHWC pad-1 patches in (ky,kx,c) order x packed-W^T == conv2d.
    torch.manual_seed(2)
    n, ci, hw, co = 2, 16, 7, 32
    x = torch.randn(n, ci, hw, hw)
    w = torch.randn(co, ci, 3, 3)
    b = torch.randn(co)
    p = _pack_resnet_weight(w).float()
    xp = F.pad(x, (1, 1, 1, 1))
    patches = (
        F.unfold(xp, 3)  # [n, ci*3*3, hw*hw] in (c, ky, kx) order
        .view(n, ci, 3, 3, hw * hw)
        .permute(0, 4, 2, 3, 1)  # -> (ky, kx, c)
        .reshape(n * hw * hw, 3 * 3 * ci)
        .to(torch.bfloat16).float()
    )
    out = (patches @ p[:, :3 * 3 * ci].t() + b).view(n, hw, hw, co)
    ref = F.conv2d(x.to(torch.bfloat16).float(),
                   w.to(torch.bfloat16).float(), b, padding=1)
    torch.testing.assert_close(out.permute(0, 3, 1, 2), ref,
                               rtol=2e-2, atol=2e-2)


def test_wgrad_unpack_roundtrip():
    # The kernel returns dW as [ky][co][kx*ci+c] (K-cols padded to 16);
    # the autograd Function unpacks with view+permute. Verify against a
    # hand-built dwp from a known dw.
    torch.manual_seed(3)
    co, ci = 16, 16
    dw = torch.randn(co, ci, 3, 3)
    kwc = 3 * ci
    kwcp = (kwc + 15) // 16 * 16
    dwp = torch.zeros(3, co, kwcp)
    for ky in range(3):
        for kx in range(3):
            for c in range(ci):
                dwp[ky, :, kx * ci + c] = dw[:, c, ky, kx]
    got = (dwp[:, :, :kwc].view(3, co, 3, ci).permute(1, 3, 0, 2))
    torch.testing.assert_close(got, dw)
