"""Micro-benchmark: MFMA trunk vs MIOpen/aten on the learner batch.

Run on a GPU box:  python scripts/bench_conv_mfma.py [N]
Writes timings to stdout (fwd-only and fwd+bwd, both paths).
"""

import os
import sys
import timeit

sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))

import torch
import torch.nn.functional as F

from torchbeast_amd.ops import functional as tbf


def eager_fwd(frames, c1, c2, c3):
    x = frames.float() / 255.0
    x = F.relu(c1(x))
    x = F.relu(c2(x))
    x = F.relu(c3(x))
    return x.reshape(frames.shape[0], -1)


def timed(fn, iters=30, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = timeit.default_timer()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (timeit.default_timer() - t0) / iters * 1e3


def main():
    N = int(sys.argv[1]) if len(sys.argv) > 1 else 2592
    torch.manual_seed(0)
    dev = "cuda"
    c1 = torch.nn.Conv2d(4, 32, 8, stride=4).to(dev)
    c2 = torch.nn.Conv2d(32, 64, 4, stride=2).to(dev)
    c3 = torch.nn.Conv2d(64, 64, 3, stride=1).to(dev)
    frames = torch.randint(0, 256, (N, 4, 84, 84), dtype=torch.uint8,
                           device=dev)
    d_out = torch.randn(N, 3136, device=dev)

    def mfma_fwd():
        with torch.no_grad():
            import torchbeast_amd.ops as om
            ext = om.require_ext()
            w1p, w2p, w3p = tbf._pack_trunk_weights(c1.weight, c2.weight,
                                                    c3.weight)
            ext.conv_trunk_fwd(frames, w1p, c1.bias.detach().contiguous(),
                               w2p, c2.bias.detach().contiguous(), w3p,
                               c3.bias.detach().contiguous(), False)

    def mfma_fwd_bwd():
        for c in (c1, c2, c3):
            c.weight.grad = None
            c.bias.grad = None
        out = tbf._AtariTrunkMfma.apply(frames, c1.weight, c1.bias,
                                        c2.weight, c2.bias, c3.weight,
                                        c3.bias)
        out.backward(d_out)

    def lib_fwd():
        with torch.no_grad():
            eager_fwd(frames, c1, c2, c3)

    def lib_fwd_bwd():
        for c in (c1, c2, c3):
            c.weight.grad = None
            c.bias.grad = None
        out = eager_fwd(frames, c1, c2, c3)
        out.backward(d_out)

    print(f"N={N}")
    print(f"mfma fwd      : {timed(mfma_fwd):8.3f} ms")
    print(f"lib  fwd      : {timed(lib_fwd):8.3f} ms")
    print(f"mfma fwd+bwd  : {timed(mfma_fwd_bwd):8.3f} ms")
    print(f"lib  fwd+bwd  : {timed(lib_fwd_bwd):8.3f} ms")


if __name__ == "__main__":
    main()
