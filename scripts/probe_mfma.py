import os, sys
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import torch, timeit
import torchbeast_amd.ops as _ops
_tbops = _ops.require_ext()
torch.manual_seed(0)
M,N,K = 128, 128, 64
A = torch.randn(M,K,device="cuda").bfloat16()
B = torch.randn(N,K,device="cuda").bfloat16()
ref = (A.float() @ B.float().t())
matches = []
for v in range(4):
    C = _tbops.mfma_gemm_probe(A,B,v)
    rel = ((C-ref).abs().max() / ref.abs().max()).item()
    tag = "MATCH" if rel < 0.02 else "no"
    if rel < 0.02: matches.append(v)
    print("variant", v, "rel_err", round(rel,5), tag)
for v in matches:
    A2 = torch.randn(2048,2048,device="cuda").bfloat16(); B2 = torch.randn(2048,2048,device="cuda").bfloat16()
    _tbops.mfma_gemm_probe(A2,B2,v); torch.cuda.synchronize()
    t0=timeit.default_timer()
    for _ in range(20): _tbops.mfma_gemm_probe(A2,B2,v)
    torch.cuda.synchronize()
    dt=(timeit.default_timer()-t0)/20
    print("variant", v, "2048^3:", round(2*2048**3/dt/1e12,1), "TFLOP/s")
    A4 = torch.randn(4096,4096,device="cuda").bfloat16(); B4 = torch.randn(4096,4096,device="cuda").bfloat16()
    _tbops.mfma_gemm_probe(A4,B4,v); torch.cuda.synchronize()
    t0=timeit.default_timer()
    for _ in range(10): _tbops.mfma_gemm_probe(A4,B4,v)
    torch.cuda.synchronize()
    dt=(timeit.default_timer()-t0)/10
    print("variant", v, "4096^3:", round(2*4096**3/dt/1e12,1), "TFLOP/s")
# hipblaslt reference
A4 = torch.randn(4096,4096,device="cuda").bfloat16(); B4 = torch.randn(4096,4096,device="cuda").bfloat16()
torch.matmul(A4, B4.t()); torch.cuda.synchronize()
t0=timeit.default_timer()
for _ in range(10): torch.matmul(A4, B4.t())
torch.cuda.synchronize()
dt=(timeit.default_timer()-t0)/10
print("hipblaslt bf16 4096^3:", round(2*4096**3/dt/1e12,1), "TFLOP/s")
