"""Synthetic environments (no gym dependency).

`SyntheticAtariEnv` generalizes the reference's built-in "Mock" env
(ref: torchbeast/polybeast_env.py:39-46): Atari-shaped uint8 frames with a
cheap deterministic generator, configurable episode length/reward pattern.
It is the benchmark environment (BASELINE.json configs 2-5: "synthetic
frames, random-init weights") and speaks the classic gym step API so it can
sit behind `core.environment.Environment`, the C++ env server, or the
in-process C++ actor fast path (which reimplements the same generator
natively; see runtime/csrc/synth_env.h).

`CountingEnv` is the deterministic fixture for recurrent-agent-state tests
(ref: tests/core_agent_state_env.py): observation = step counter, episodes
are `episode_length` steps, reward equals the counter so tests can check
bookkeeping exactly.
"""

import numpy as np


class SyntheticAtariEnv:
    """Atari-shaped synthetic env: uint8 frames from an xorshift generator."""

    def __init__(
        self,
        shape=(4, 84, 84),
        num_actions=6,
        episode_length=1000,
        seed=1,
    ):
        self.shape = tuple(shape)
        self.num_actions = num_actions
        self.episode_length = episode_length
        self._state = np.uint64(seed * 2654435761 + 1)
        self._t = 0

    def _next_frame(self):
        # xorshift64* — one scalar update per frame, then broadcast into a
        # pattern; cheap enough that the env never bounds the pipeline.
        s = self._state
        s ^= s << np.uint64(13)
        s ^= s >> np.uint64(7)
        s ^= s << np.uint64(17)
        self._state = s
        base = np.uint8(s & np.uint64(0xFF))
        frame = np.empty(self.shape, dtype=np.uint8)
        frame.fill(base)
        # Cheap spatial variation so frames aren't constant.
        frame[..., :: 8] ^= np.uint8((s >> np.uint64(8)) & np.uint64(0xFF))
        return frame

    def reset(self):
        self._t = 0
        return self._next_frame()

    def step(self, action):
        self._t += 1
        done = self._t >= self.episode_length
        # Deterministic pseudo-reward in {-1, 0, 1}.
        reward = float((int(self._state) + int(action)) % 3 - 1)
        return self._next_frame(), reward, done, {}

    def close(self):
        pass


class BanditEnv:
    """Learnability fixture: constant observation, reward 1 for the target
    action and 0 otherwise. A correct IMPALA implementation drives the
    policy toward the target within a few thousand steps; tests assert the
    mean episode return rises (the framework-level analogue of the
    reference's 'learning curves equivalent' claim)."""

    def __init__(self, shape=(4, 36, 36), num_actions=4, target_action=2,
                 episode_length=20):
        self.shape = tuple(shape)
        self.num_actions = num_actions
        self.target_action = target_action
        self.episode_length = episode_length
        self._t = 0
        self._obs = np.full(self.shape, 128, dtype=np.uint8)

    def reset(self):
        self._t = 0
        return self._obs

    def step(self, action):
        self._t += 1
        reward = 1.0 if int(action) == self.target_action else 0.0
        done = self._t >= self.episode_length
        if done:
            self._t = 0
        return self._obs, reward, done, {}

    def close(self):
        pass


class CountingEnv:
    """Observation = global step counter; episodes of fixed length.

    The counter never resets across episodes, which lets tests assert that
    rollout boundaries, auto-resets and recurrent-state resets line up with
    the exact frame index (ref: tests/core_agent_state_env.py:21-34).
    """

    def __init__(self, shape=(1,), episode_length=5, dtype=np.float32):
        self.shape = tuple(shape)
        self.episode_length = episode_length
        self.dtype = dtype
        self._counter = 0

    def _obs(self):
        return np.full(self.shape, self._counter, dtype=self.dtype)

    def reset(self):
        return self._obs()

    def step(self, action):
        self._counter += 1
        done = self._counter % self.episode_length == 0
        reward = float(self._counter)
        return self._obs(), reward, done, {}

    def close(self):
        pass
