import os, sys, timeit
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch
import torchbeast_amd.ops as _ops
ext = _ops.require_ext()
torch.manual_seed(0)
A = torch.randn(256, 128, device="cuda").bfloat16()
B = torch.randn(128, 128, device="cuda").bfloat16()
ref = A.float() @ B.float().t()
C = ext.mfma_gemm_v2(A, B)
rel = ((C - ref).abs().max() / ref.abs().max()).item()
print("v2 numerics rel_err:", round(rel, 6), "MATCH" if rel < 0.02 else "MISMATCH")
if rel < 0.02:
    for n in (2048, 4096):
        A2 = torch.randn(n, n, device="cuda").bfloat16()
        B2 = torch.randn(n, n, device="cuda").bfloat16()
        ext.mfma_gemm_v2(A2, B2); torch.cuda.synchronize()
        t0 = timeit.default_timer()
        iters = 20 if n == 2048 else 10
        for _ in range(iters): ext.mfma_gemm_v2(A2, B2)
        torch.cuda.synchronize()
        dt = (timeit.default_timer() - t0) / iters
        print(f"v2 {n}^3:", round(2 * n**3 / dt / 1e12, 1), "TFLOP/s")
