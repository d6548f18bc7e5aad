"""Microbenchmark: fused AtariNet conv trunk vs MIOpen eager chain.

GPU-only. Compares forward (inference + learner batch sizes) and the full
fwd+bwd learner path. Run under gpurun; prints one line per config.
"""

import os
import sys
import timeit

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from torchbeast_amd.ops import functional as tbops


def time_fn(fn, iters=50, warmup=10):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = timeit.default_timer()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (timeit.default_timer() - t0) / iters * 1e3  # ms


def main():
    assert torch.cuda.is_available()
    conv1 = torch.nn.Conv2d(4, 32, 8, stride=4).cuda()
    conv2 = torch.nn.Conv2d(32, 64, 4, stride=2).cuda()
    conv3 = torch.nn.Conv2d(64, 64, 3, stride=1).cuda()

    def eager_fwd(x):
        return F.relu(conv3(F.relu(conv2(F.relu(conv1(x)))))).flatten(1)

    for N in (64, 128, 256, 512, 2592):
        frames = torch.randint(0, 256, (N, 4, 84, 84), dtype=torch.uint8,
                               device="cuda")
        frames_f = frames.float() / 255.0

        with torch.no_grad():
            t_fused = time_fn(
                lambda: tbops.atari_trunk(frames, conv1, conv2, conv3))
            t_eager = time_fn(lambda: eager_fwd(frames_f))
            t_eager_u8 = time_fn(lambda: eager_fwd(frames.float() / 255.0))
        print(f"N={N:5d} fwd: fused={t_fused:7.3f} ms  eager={t_eager:7.3f} ms"
              f"  eager+cvt={t_eager_u8:7.3f} ms", flush=True)

        if N == 2592:
            def fused_step():
                out = tbops.atari_trunk(frames, conv1, conv2, conv3)
                out.square().sum().backward()
                for c in (conv1, conv2, conv3):
                    c.weight.grad = None
                    c.bias.grad = None

            def eager_step():
                out = eager_fwd(frames.float() / 255.0)
                out.square().sum().backward()
                for c in (conv1, conv2, conv3):
                    c.weight.grad = None
                    c.bias.grad = None

            t_fused_b = time_fn(fused_step, iters=20)
            t_eager_b = time_fn(eager_step, iters=20)
            print(f"N={N:5d} fwd+bwd: fused={t_fused_b:7.3f} ms  "
                  f"eager={t_eager_b:7.3f} ms", flush=True)


if __name__ == "__main__":
    main()
