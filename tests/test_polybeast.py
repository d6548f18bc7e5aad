"""PolyBeast end-to-end on CPU: in-process synthetic envs and socket env
servers; checkpoint write + resume."""

import os
import uuid

import torch

from torchbeast_amd import polybeast_learner


def _flags(tmp_path, **overrides):
    flags = polybeast_learner.parser.parse_args([])
    flags.env = "synthetic:4x36x36:6"
    flags.savedir = str(tmp_path)
    flags.xpid = "pbtest"
    flags.num_actors = 4
    flags.batch_size = 2
    flags.unroll_length = 8
    flags.total_steps = 64
    flags.num_learner_threads = 1
    flags.num_inference_threads = 1
    flags.disable_cuda = True
    for k, v in overrides.items():
        setattr(flags, k, v)
    return flags


def test_train_synthetic_inproc(tmp_path):
    flags = _flags(tmp_path)
    polybeast_learner.train(flags)
    ckpt = torch.load(
        os.path.join(str(tmp_path), "pbtest", "model.tar"),
        map_location="cpu",
        weights_only=False,
    )
    assert set(ckpt) >= {
        "model_state_dict",
        "optimizer_state_dict",
        "scheduler_state_dict",
        "stats",
        "flags",
    }
    assert ckpt["stats"]["step"] >= 64


def test_train_resumes_from_checkpoint(tmp_path):
    flags = _flags(tmp_path)
    polybeast_learner.train(flags)
    # Second run resumes and extends.
    flags2 = _flags(tmp_path, total_steps=128)
    polybeast_learner.train(flags2)
    ckpt = torch.load(
        os.path.join(str(tmp_path), "pbtest", "model.tar"),
        map_location="cpu",
        weights_only=False,
    )
    assert ckpt["stats"]["step"] >= 128


def test_train_with_lstm(tmp_path):
    flags = _flags(tmp_path, use_lstm=True, total_steps=32)
    polybeast_learner.train(flags)
    assert os.path.exists(os.path.join(str(tmp_path), "pbtest", "model.tar"))


def test_train_deep_model(tmp_path):
    flags = _flags(tmp_path, model="deep", total_steps=32,
                   env="synthetic:4x84x84:6")
    polybeast_learner.train(flags)


def test_train_against_socket_env_servers(tmp_path):
    """Non-synthetic env name -> the learner connects to unix-socket env
    servers at pipes_basename.{i} (here: servers hosting the Python
    synthetic env, matching the learner's default 84x84x4 model shape)."""
    from torchbeast_amd import polybeast_env, runtime

    basename = f"unix:/tmp/tbamd-pb-{uuid.uuid4().hex[:8]}"
    factory = polybeast_env.create_env_factory("synthetic:4x84x84:6")
    servers = [runtime.Server(factory, f"{basename}.{i}") for i in range(2)]
    for s in servers:
        s.start()
    try:
        flags = _flags(tmp_path, num_actors=2, pipes_basename=basename,
                       total_steps=32)
        flags.env = "PongNoFrameskip-v4"  # non-synthetic -> socket path
        flags.num_actions = 6
        polybeast_learner.train(flags)
    finally:
        for s in servers:
            s.stop()


def test_combined_cli_launcher(tmp_path):
    """polybeast.py combined CLI (ref: torchbeast/polybeast.py) end-to-end
    as a subprocess with in-process synthetic envs."""
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, "-m", "torchbeast_amd.polybeast",
         "--env", "synthetic:4x36x36:6", "--num_actors", "2",
         "--batch_size", "2", "--unroll_length", "8", "--total_steps", "32",
         "--num_learner_threads", "1", "--num_inference_threads", "1",
         "--disable_cuda", "--savedir", str(tmp_path), "--xpid", "combined"],
        cwd=os.path.dirname(os.path.dirname(os.path.abspath(__file__))),
        timeout=240,
        capture_output=True,
    )
    assert out.returncode == 0, out.stderr.decode()[-2000:]
    assert os.path.exists(os.path.join(str(tmp_path), "combined", "model.tar"))


def test_polybeast_test_mode(tmp_path):
    """test mode loads the checkpoint and runs greedy local-env episodes."""
    flags = _flags(tmp_path, total_steps=32)
    polybeast_learner.train(flags)
    avg = polybeast_learner.test(flags, num_episodes=1)
    assert isinstance(avg, float)


def test_train_two_learner_threads(tmp_path):
    """Two learner threads share the model under the learn lock (the
    reference's default num_learner_threads=2 topology)."""
    flags = _flags(tmp_path, num_learner_threads=2, total_steps=96)
    polybeast_learner.train(flags)
    ckpt = torch.load(
        os.path.join(str(tmp_path), "pbtest", "model.tar"),
        map_location="cpu", weights_only=False,
    )
    assert ckpt["stats"]["step"] >= 96


def test_train_lstm_over_socket_servers(tmp_path):
    """Recurrent state round-trips through the wire protocol + batcher while
    training against env servers."""
    from torchbeast_amd import polybeast_env, runtime

    basename = f"unix:/tmp/tbamd-pbl-{uuid.uuid4().hex[:8]}"
    factory = polybeast_env.create_env_factory("synthetic:4x84x84:6")
    servers = [runtime.Server(factory, f"{basename}.{i}") for i in range(2)]
    for s in servers:
        s.start()
    try:
        flags = _flags(tmp_path, num_actors=2, pipes_basename=basename,
                       total_steps=32, use_lstm=True)
        flags.env = "PongNoFrameskip-v4"  # socket addressing path
        flags.num_actions = 6
        polybeast_learner.train(flags)
    finally:
        for s in servers:
            s.stop()
