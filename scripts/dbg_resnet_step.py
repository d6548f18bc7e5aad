"""Time the deep-ResNet trunk fwd/bwd per conv size, MFMA vs aten."""
import os
import sys
import time

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch  # noqa: E402
import torch.nn.functional as F  # noqa: E402

from torchbeast_amd.models.resnet import ResNet  # noqa: E402
from torchbeast_amd.ops import functional as tbf  # noqa: E402

N = 2560  # learner flat batch (T*B = 80*32)


def timeit(fn, iters=8):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    torch.manual_seed(0)
    net = ResNet((4, 84, 84), 6).cuda()
    x = torch.randn(N, 4, 84, 84, device="cuda")

    for mode in ("mfma", "aten"):
        os.environ["TBAMD_RESNET"] = mode
        def step():
            out = net._features(x)
            out.square().mean().backward()
            net.zero_grad(set_to_none=True)
        ms = timeit(step)
        print(f"trunk fwd+bwd [{mode}]: {ms:.2f} ms")

    # Per-geometry single conv cost, fwd and fwd+bwd.
    for ci, hw, co in [(16, 42, 16), (16, 42, 32), (32, 21, 32),
                       (32, 11, 32)]:
        conv = torch.nn.Conv2d(ci, co, 3, padding=1).cuda()
        xb = (torch.randn(N, ci, hw, hw, device="cuda")
              .to(torch.bfloat16)
              .contiguous(memory_format=torch.channels_last)
              .requires_grad_(True))
        with torch.no_grad():
            f_ms = timeit(lambda: tbf.resnet_conv3x3(conv, xb))
        def fb():
            out = tbf.resnet_conv3x3(conv, xb)
            out.float().square().mean().backward()
            conv.zero_grad(set_to_none=True)
            xb.grad = None
        fb_ms = timeit(fb)
        # aten bf16 channels_last comparison
        convb = torch.nn.Conv2d(ci, co, 3, padding=1).cuda().to(
            torch.bfloat16).to(memory_format=torch.channels_last)
        with torch.no_grad():
            a_ms = timeit(lambda: F.conv2d(xb, convb.weight, convb.bias,
                                           padding=1))
        def afb():
            out = F.conv2d(xb, convb.weight, convb.bias, padding=1)
            out.float().square().mean().backward()
            convb.zero_grad(set_to_none=True)
            xb.grad = None
        afb_ms = timeit(afb)
        print(f"conv {ci:2d}->{co:2d} @{hw:2d}: mfma fwd {f_ms:6.2f} "
              f"fwd+bwd {fb_ms:6.2f} | aten-bf16 fwd {a_ms:6.2f} "
              f"fwd+bwd {afb_ms:6.2f} ms")


if __name__ == "__main__":
    main()
