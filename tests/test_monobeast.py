"""End-to-end monobeast smoke: a tiny synthetic-env training run on CPU,
plus checkpoint save/load and the test-mode path (ref test strategy:
function-level integration)."""

import os

import torch

from torchbeast_amd import monobeast


def _tiny_flags(tmp_path, **overrides):
    flags = monobeast.parser.parse_args([])
    flags.env = "synthetic:4x36x36:6"
    flags.savedir = str(tmp_path)
    flags.xpid = "smoke"
    flags.num_actors = 2
    flags.num_buffers = 6
    flags.batch_size = 2
    flags.unroll_length = 8
    flags.total_steps = 128
    flags.num_learner_threads = 1
    flags.disable_cuda = True
    for k, v in overrides.items():
        setattr(flags, k, v)
    return flags


def test_train_smoke_and_checkpoint(tmp_path):
    flags = _tiny_flags(tmp_path)
    monobeast.train(flags)

    ckpt_path = os.path.join(str(tmp_path), "smoke", "model.tar")
    assert os.path.exists(ckpt_path)
    ckpt = torch.load(ckpt_path, map_location="cpu", weights_only=False)
    assert set(ckpt) >= {
        "model_state_dict",
        "optimizer_state_dict",
        "scheduler_state_dict",
        "flags",
    }

    # Test mode loads the checkpoint and runs greedy episodes.
    flags.mode = "test"
    monobeast.test(flags, num_episodes=1)


def test_train_smoke_with_lstm(tmp_path):
    flags = _tiny_flags(tmp_path, use_lstm=True, total_steps=64)
    monobeast.train(flags)
    assert os.path.exists(os.path.join(str(tmp_path), "smoke", "model.tar"))


def test_learn_updates_weights(tmp_path):
    """learn() on a synthetic batch must change the learner parameters and
    sync them into the actor model."""
    flags = _tiny_flags(tmp_path)
    flags.device = torch.device("cpu")
    T, B = flags.unroll_length, flags.batch_size
    shape, A = (4, 36, 36), 6

    model = monobeast.Net(shape, A)
    actor_model = monobeast.Net(shape, A)
    optimizer = torch.optim.RMSprop(model.parameters(), lr=0.01)
    scheduler = torch.optim.lr_scheduler.LambdaLR(optimizer, lambda e: 1.0)

    batch = dict(
        frame=torch.randint(0, 255, (T + 1, B, *shape), dtype=torch.uint8),
        reward=torch.randn(T + 1, B),
        done=torch.rand(T + 1, B) < 0.05,
        episode_return=torch.randn(T + 1, B),
        episode_step=torch.randint(0, 100, (T + 1, B), dtype=torch.int32),
        last_action=torch.randint(0, A, (T + 1, B)),
        action=torch.randint(0, A, (T + 1, B)),
        policy_logits=torch.randn(T + 1, B, A),
        baseline=torch.randn(T + 1, B),
    )

    before = [p.clone() for p in model.parameters()]
    stats = monobeast.learn(
        flags, actor_model, model, batch, (), optimizer, scheduler
    )
    assert "total_loss" in stats and "pg_loss" in stats
    changed = any(
        not torch.equal(b, p) for b, p in zip(before, model.parameters())
    )
    assert changed
    for p_actor, p_model in zip(actor_model.parameters(), model.parameters()):
        torch.testing.assert_close(p_actor, p_model)
