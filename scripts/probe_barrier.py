"""Barrier/launch-overhead probes for the LSTM redesign. Run on a GPU box."""
import os, sys
sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
import torch
import torchbeast_amd.ops as om

ext = om.require_ext()
print(f"empty-kernel launch: {om.require_ext().launch_probe(2000)*1e6:.2f} us")
ITERS = 2000
for wgs in (8, 16, 32, 64, 128, 256):
    row = [f"wgs={wgs:4d}"]
    for var, name in ((0, "cg"), (1, "fence"), (2, "scoped")):
        secs, got, exp = ext.barrier_probe(var, wgs, ITERS)
        us = secs / (2 * ITERS) * 1e6  # 2 barriers per iter
        ok = "ok" if got == exp else f"BAD({got}!={exp})"
        row.append(f"{name}={us:7.3f}us {ok}")
    print("  ".join(row))
