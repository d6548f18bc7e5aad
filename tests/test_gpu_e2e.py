"""End-to-end GPU tests: a short polybeast training run and a mini bench."""

import json
import os
import subprocess
import sys

import pytest
import torch

pytestmark = pytest.mark.gpu

if not torch.cuda.is_available():  # pragma: no cover
    pytest.skip("requires ROCm GPU", allow_module_level=True)


def test_polybeast_train_short_gpu(tmp_path):
    from torchbeast_amd import polybeast_learner

    flags = polybeast_learner.parser.parse_args([])
    flags.env = "synthetic:4x84x84:6"
    flags.savedir = str(tmp_path)
    flags.xpid = "gpue2e"
    flags.num_actors = 16
    flags.batch_size = 8
    flags.unroll_length = 20
    flags.total_steps = 8 * 20 * 6
    flags.num_learner_threads = 1
    flags.num_inference_threads = 1
    polybeast_learner.train(flags)
    assert os.path.exists(os.path.join(str(tmp_path), "gpue2e", "model.tar"))


def test_bench_smoke():
    out = subprocess.check_output(
        [sys.executable, "bench.py", "--gpus", "1", "--steps", "5",
         "--warmup", "3", "--batch_size", "8", "--actors", "16"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        timeout=600,
    )
    line = out.decode().strip().splitlines()[-1]
    result = json.loads(line)
    assert result["value"] > 0
    assert result["n_gpus"] == 1
    assert result["data"] == "synthetic"


def test_graft_smoke():
    sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
    import __graft_entry__

    __graft_entry__.smoke()


def test_monobeast_train_short_gpu(tmp_path):
    """Monobeast on GPU: CPU actor processes + shared-memory buffers feed a
    GPU learner running the fused HIP ops."""
    from torchbeast_amd import monobeast

    flags = monobeast.parser.parse_args([])
    flags.env = "synthetic:4x84x84:6"
    flags.savedir = str(tmp_path)
    flags.xpid = "monogpu"
    flags.num_actors = 2
    flags.num_buffers = 6
    flags.batch_size = 2
    flags.unroll_length = 16
    flags.total_steps = 16 * 2 * 5
    flags.num_learner_threads = 1
    monobeast.train(flags)
    assert os.path.exists(os.path.join(str(tmp_path), "monogpu", "model.tar"))
