"""Isolated inference-serve-path timings (trunk variants + packing)."""
import os, sys, timeit
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import torchbeast_amd.ops as om

ext = om.require_ext()
torch.manual_seed(0)
dev = "cuda"
w1 = torch.randn(32, 4, 8, 8, device=dev); b1 = torch.randn(32, device=dev)
w2 = torch.randn(64, 32, 4, 4, device=dev); b2 = torch.randn(64, device=dev)
w3 = torch.randn(64, 64, 3, 3, device=dev); b3 = torch.randn(64, device=dev)

def timed(fn, iters=50, warmup=10):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = timeit.default_timer()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (timeit.default_timer() - t0) / iters * 1e3

for N in (64, 128, 256, 512):
    frames = torch.randint(0, 256, (N, 4, 84, 84), dtype=torch.uint8, device=dev)
    def mfma_with_pack():
        w1p = w1.reshape(32, -1).to(torch.bfloat16).contiguous()
        w2p = w2.permute(0, 2, 3, 1).reshape(64, -1).to(torch.bfloat16).contiguous()
        w3p = w3.permute(0, 2, 3, 1).reshape(64, -1).to(torch.bfloat16).contiguous()
        ext.conv_trunk_fwd(frames, w1p, b1, w2p, b2, w3p, b3, False)
    w1p = w1.reshape(32, -1).to(torch.bfloat16).contiguous()
    w2p = w2.permute(0, 2, 3, 1).reshape(64, -1).to(torch.bfloat16).contiguous()
    w3p = w3.permute(0, 2, 3, 1).reshape(64, -1).to(torch.bfloat16).contiguous()
    def mfma_pre():
        ext.conv_trunk_fwd(frames, w1p, b1, w2p, b2, w3p, b3, False)
    def valu():
        ext.atari_trunk_fwd(frames, w1, b1, w2, b2, w3, b3, False)
    print(f"N={N:4d}  mfma+pack={timed(mfma_with_pack):7.3f}ms  "
          f"mfma={timed(mfma_pre):7.3f}ms  valu={timed(valu):7.3f}ms")
