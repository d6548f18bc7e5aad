"""End-to-end learning check: the full polybeast stack (C++ actor pool,
batching, V-trace, fused losses, fused RMSProp) must actually improve the
policy on a trivially learnable bandit environment. This is the
framework-level analogue of the reference's 'learning curves equivalent to
TF IMPALA' qualitative claim (ref: README.md:219-229)."""

import threading
import uuid

import pytest
import torch

from torchbeast_amd import polybeast_learner as pbl
from torchbeast_amd import runtime
from torchbeast_amd.envs.synthetic import BanditEnv


@pytest.mark.timeout(300)
def test_polybeast_learns_bandit(tmp_path):
    basename = f"unix:/tmp/tbamd-learn-{uuid.uuid4().hex[:8]}"
    servers = []
    for i in range(4):
        server = runtime.Server(
            lambda: BanditEnv(shape=(4, 36, 36), num_actions=4,
                              target_action=2, episode_length=20),
            f"{basename}.{i}",
        )
        server.start()
        servers.append(server)

    flags = pbl.parser.parse_args([])
    flags.env = "bandit"  # non-synthetic name -> socket addresses
    flags.pipes_basename = basename
    flags.savedir = str(tmp_path)
    flags.xpid = "learntest"
    flags.disable_checkpoint = True
    flags.num_actors = 4
    flags.num_actions = 4
    flags.batch_size = 4
    flags.unroll_length = 20
    flags.total_steps = 24000
    flags.num_learner_threads = 1
    flags.num_inference_threads = 1
    flags.disable_cuda = True
    flags.learning_rate = 0.005
    flags.entropy_cost = 0.01

    # The learner's default model expects 84x84; match the env's 36x36 by
    # masquerading as a synthetic spec for model construction only.
    import torchbeast_amd.flags as tbflags

    orig = tbflags.parse_synthetic_env_spec

    def spec(env):
        if env == "bandit":
            return None  # socket path for addressing...
        return orig(env)

    # Simplest: monkeypatch observation_shape.
    orig_shape = pbl.observation_shape
    pbl.observation_shape = lambda f: ((4, 36, 36), 4)
    try:
        stats = {}

        def run():
            pbl.train(flags)

        t = threading.Thread(target=run)
        t.start()
        t.join(240)
        assert not t.is_alive(), "training did not finish"
    finally:
        pbl.observation_shape = orig_shape
        for s in servers:
            s.stop()

    ckpt_dir = tmp_path / "learntest"
    # Read the logged returns: early vs late mean_episode_return.
    import csv

    rows = []
    with open(ckpt_dir / "logs.csv") as f:
        lines = f.readlines()
    fieldnames = lines[0].lstrip("# ").strip().split(",")
    for row in csv.DictReader((l for l in lines[1:] if not l.startswith("#")),
                              fieldnames=fieldnames):
        v = row.get("mean_episode_return")
        if v not in (None, "", "None", "nan"):
            rows.append(float(v))
    assert len(rows) >= 10, f"too few return samples: {len(rows)}"
    early = sum(rows[:5]) / 5
    late = sum(rows[-5:]) / 5
    # Random policy: ~5 of 20 steps hit the target (E[return]=5); learned
    # policy approaches 20. Require clear improvement and near-optimality.
    assert late > early + 3, f"no learning: early={early:.2f} late={late:.2f}"
    assert late > 12, f"policy far from optimal: late={late:.2f}"


@pytest.mark.gpu
@pytest.mark.timeout(420)
def test_polybeast_learns_bandit_gpu_bf16_path(tmp_path):
    """Same learnability check but ON the GPU with 84x84 frames, so the
    bandit gradients flow through the bf16 MFMA trunk, the fused losses,
    V-trace and the fused RMSProp — evidence that the mixed-precision
    training path actually learns, not just that its numerics match an
    oracle pointwise."""
    if not torch.cuda.is_available():
        pytest.skip("requires ROCm GPU")

    basename = f"unix:/tmp/tbamd-learng-{uuid.uuid4().hex[:8]}"
    servers = []
    for i in range(8):
        server = runtime.Server(
            lambda: BanditEnv(shape=(4, 84, 84), num_actions=4,
                              target_action=2, episode_length=20),
            f"{basename}.{i}",
        )
        server.start()
        servers.append(server)

    flags = pbl.parser.parse_args([])
    flags.env = "bandit"
    flags.pipes_basename = basename
    flags.savedir = str(tmp_path)
    flags.xpid = "learngpu"
    flags.disable_checkpoint = True
    flags.num_actors = 8
    flags.num_actions = 4
    flags.batch_size = 8
    flags.unroll_length = 20
    flags.total_steps = 48000
    flags.num_learner_threads = 1
    flags.num_inference_threads = 1
    flags.learning_rate = 0.005
    flags.entropy_cost = 0.01

    orig_shape = pbl.observation_shape
    pbl.observation_shape = lambda f: ((4, 84, 84), 4)
    try:
        t = threading.Thread(target=lambda: pbl.train(flags))
        t.start()
        t.join(360)
        assert not t.is_alive(), "training did not finish"
    finally:
        pbl.observation_shape = orig_shape
        for s in servers:
            s.stop()

    import csv

    rows = []
    with open(tmp_path / "learngpu" / "logs.csv") as f:
        lines = f.readlines()
    fieldnames = lines[0].lstrip("# ").strip().split(",")
    for row in csv.DictReader((l for l in lines[1:] if not l.startswith("#")),
                              fieldnames=fieldnames):
        v = row.get("mean_episode_return")
        if v not in (None, "", "None", "nan"):
            rows.append(float(v))
    assert len(rows) >= 6, f"too few return samples: {len(rows)}"
    early = sum(rows[:3]) / 3
    late = sum(rows[-3:]) / 3
    assert late > early + 3, f"no learning: early={early:.2f} late={late:.2f}"
    assert late > 12, f"policy far from optimal: late={late:.2f}"
