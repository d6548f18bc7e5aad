"""World-2 torchrun of bench.py itself on CPU/gloo: the exact command
shape the driver uses for the round-end SCALE run (one rank per GPU; here
gloo ranks on CPU). Checks rank 0 emits the one-line JSON contract with
whole-job aggregation."""
import json
import os
import socket
import subprocess
import sys

import pytest


def _free_port():
    s = socket.socket()
    s.bind(("127.0.0.1", 0))
    port = s.getsockname()[1]
    s.close()
    return port


@pytest.mark.timeout(600)
def test_bench_world2_cpu_contract():
    root = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
    env = dict(os.environ)
    env.pop("RANK", None)
    env.pop("WORLD_SIZE", None)
    out = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", str(_free_port()), "bench.py",
         "--gpus", "2", "--steps", "2", "--warmup", "1",
         "--actors", "8", "--batch_size", "4", "--unroll_length", "16"],
        cwd=root, env=env, timeout=540, capture_output=True,
    )
    assert out.returncode == 0, out.stderr.decode()[-4000:]
    lines = [ln for ln in out.stdout.decode().splitlines()
             if ln.startswith("{")]
    assert len(lines) == 1, f"rank 0 must print exactly one JSON line: {lines}"
    d = json.loads(lines[0])
    assert d["n_gpus"] == 2
    assert d["steps"] == 2 and d["warmup"] == 1
    assert d["value"] > 0 and d["ms_per_step"] > 0
    assert d["scaling"] == "weak"
    assert d["data"] == "synthetic"
