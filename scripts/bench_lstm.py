"""Isolated LSTM unroll timings: fused kernel vs per-phase breakdown.

python scripts/bench_lstm.py [T B H L]
"""
import os, sys, timeit
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
from torchbeast_amd.ops import functional as tbf

T = int(sys.argv[1]) if len(sys.argv) > 1 else 80
B = int(sys.argv[2]) if len(sys.argv) > 2 else 32
H = int(sys.argv[3]) if len(sys.argv) > 3 else 519
L = int(sys.argv[4]) if len(sys.argv) > 4 else 2

torch.manual_seed(0)
dev = "cuda"
core = torch.nn.LSTM(H, H, num_layers=L).to(dev)
x = torch.randn(T, B, H, device=dev)
notdone = (torch.rand(T, B, device=dev) > 0.05).float()
h0 = torch.zeros(L, B, H, device=dev)
c0 = torch.zeros(L, B, H, device=dev)
dout = torch.randn(T, B, H, device=dev)

def timed(fn, iters=20, warmup=5):
    for _ in range(warmup): fn()
    torch.cuda.synchronize()
    t0 = timeit.default_timer()
    for _ in range(iters): fn()
    torch.cuda.synchronize()
    return (timeit.default_timer() - t0) / iters * 1e3

def fwd():
    with torch.no_grad():
        tbf.lstm_unroll(core, x, notdone, (h0, c0))

def fwd_bwd():
    for p in core.parameters():
        p.grad = None
    out, _ = tbf.lstm_unroll(core, x, notdone, (h0, c0))
    out.backward(dout)

print(f"T={T} B={B} H={H} L={L}")
print(f"lstm fwd      : {timed(fwd):8.3f} ms")
print(f"lstm fwd+bwd  : {timed(fwd_bwd):8.3f} ms")

# Eager torch baseline (cuDNN-style per-step loop, masked).
def eager():
    state = (h0, c0)
    outs = []
    with torch.no_grad():
        for xt, nd in zip(x.unbind(), notdone.unbind()):
            nd = nd.view(1, -1, 1)
            state = tuple(nd * s for s in state)
            o, state = core(xt.unsqueeze(0), state)
            outs.append(o)

print(f"eager step-loop fwd: {timed(eager, iters=5, warmup=2):8.3f} ms")
