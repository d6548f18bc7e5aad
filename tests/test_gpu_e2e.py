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


def _repo_root():
    return os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_polybeast_train_short_gpu(tmp_path):
    # Subprocess: a fresh HIP context per training run (the long pytest
    # process accumulates CUDA state that makes in-process training runs
    # flaky on shared boxes).
    subprocess.check_call(
        [sys.executable, "-m", "torchbeast_amd.polybeast_learner",
         "--env", "synthetic:4x84x84:6", "--savedir", str(tmp_path),
         "--xpid", "gpue2e", "--num_actors", "16", "--batch_size", "8",
         "--unroll_length", "20", "--total_steps", str(8 * 20 * 6),
         "--num_learner_threads", "1", "--num_inference_threads", "1"],
        cwd=_repo_root(), timeout=300,
    )
    assert os.path.exists(os.path.join(str(tmp_path), "gpue2e", "model.tar"))


def test_polybeast_train_deep_short_gpu(tmp_path):
    # Deep IMPALA-ResNet path: MFMA 3x3 convs in both the learner autograd
    # graph and the C++ runner's serving trunk.
    subprocess.check_call(
        [sys.executable, "-m", "torchbeast_amd.polybeast_learner",
         "--env", "synthetic:4x84x84:6", "--model", "deep",
         "--savedir", str(tmp_path),
         "--xpid", "gpudeep", "--num_actors", "16", "--batch_size", "8",
         "--unroll_length", "20", "--total_steps", str(8 * 20 * 4),
         "--num_learner_threads", "1", "--num_inference_threads", "1"],
        cwd=_repo_root(), timeout=300,
    )
    assert os.path.exists(os.path.join(str(tmp_path), "gpudeep", "model.tar"))


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
    GPU learner running the fused HIP ops. Subprocess so the fork-based
    actor spawn happens in a fresh process (forking the long pytest
    process after many HIP tests is flaky)."""
    subprocess.check_call(
        [sys.executable, "-m", "torchbeast_amd.monobeast",
         "--env", "synthetic:4x84x84:6", "--savedir", str(tmp_path),
         "--xpid", "monogpu", "--num_actors", "2", "--num_buffers", "6",
         "--batch_size", "2", "--unroll_length", "16",
         "--total_steps", str(16 * 2 * 5), "--num_learner_threads", "1"],
        cwd=_repo_root(), timeout=300,
    )
    assert os.path.exists(os.path.join(str(tmp_path), "monogpu", "model.tar"))
