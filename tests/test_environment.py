"""Environment adapter + synthetic env bookkeeping."""

import numpy as np
import torch

from torchbeast_amd.core.environment import Environment
from torchbeast_amd.envs.synthetic import CountingEnv, SyntheticAtariEnv


def test_initial_protocol():
    env = Environment(CountingEnv(episode_length=5))
    out = env.initial()
    assert set(out) == {
        "frame", "reward", "done", "episode_return", "episode_step", "last_action",
    }
    assert out["frame"].shape == (1, 1, 1)
    assert out["done"].item() is True
    assert out["reward"].item() == 0.0


def test_step_and_autoreset():
    env = Environment(CountingEnv(episode_length=3))
    env.initial()
    action = torch.zeros(1, 1, dtype=torch.int64)
    for expected_counter in (1, 2):
        out = env.step(action)
        assert out["done"].item() is False
        assert out["episode_step"].item() == expected_counter
        assert out["frame"].item() == expected_counter
    out = env.step(action)
    assert out["done"].item() is True
    assert out["episode_step"].item() == 3
    assert out["episode_return"].item() == 1 + 2 + 3
    # Auto-reset: frame is from the new episode, bookkeeping was returned
    # pre-reset, and the next step starts fresh.
    out = env.step(action)
    assert out["episode_step"].item() == 1
    assert out["episode_return"].item() == 4.0


def test_synthetic_env_shapes_and_determinism():
    env_a = SyntheticAtariEnv(shape=(4, 84, 84), num_actions=6, seed=7)
    env_b = SyntheticAtariEnv(shape=(4, 84, 84), num_actions=6, seed=7)
    fa, fb = env_a.reset(), env_b.reset()
    assert fa.shape == (4, 84, 84) and fa.dtype == np.uint8
    np.testing.assert_array_equal(fa, fb)
    oa = env_a.step(3)
    ob = env_b.step(3)
    np.testing.assert_array_equal(oa[0], ob[0])
    assert oa[1] == ob[1]


def test_synthetic_env_episode_length():
    env = SyntheticAtariEnv(shape=(1, 4, 4), episode_length=4)
    env.reset()
    dones = [env.step(0)[2] for _ in range(4)]
    assert dones == [False, False, False, True]
