"""Atari preprocessing stack (ref: torchbeast/atari_wrappers.py, which
follows OpenAI baselines).

Requires `gym` (and OpenCV for frame warping); both are optional — importing
this module without them raises ImportError with a clear message, and the
rest of the framework (synthetic envs, benchmarks) does not depend on it.

Provided wrappers and builders, for capability parity with the reference:
NoopResetEnv, FireResetEnv, EpisodicLifeEnv, MaxAndSkipEnv, ClipRewardEnv,
WarpFrame, FrameStack (with LazyFrames), ScaledFloatFrame, ImageToPyTorch,
make_atari, wrap_deepmind, wrap_pytorch.
"""

import collections

import numpy as np

try:
    import gym
    from gym import spaces
except ImportError as e:  # pragma: no cover - exercised only without gym
    raise ImportError(
        "torchbeast_amd.envs.atari requires `gym` (pip install 'gym[atari]'); "
        "use torchbeast_amd.envs.synthetic for gym-free environments"
    ) from e

try:
    import cv2

    cv2.ocl.setUseOpenCL(False)
except ImportError:
    cv2 = None


class NoopResetEnv(gym.Wrapper):
    """Start each episode with a random number (1..noop_max) of no-ops."""

    def __init__(self, env, noop_max=30):
        super().__init__(env)
        self.noop_max = noop_max
        self.override_num_noops = None
        self.noop_action = 0
        assert env.unwrapped.get_action_meanings()[0] == "NOOP"

    def reset(self, **kwargs):
        self.env.reset(**kwargs)
        noops = self.override_num_noops
        if noops is None:
            noops = self.unwrapped.np_random.randint(1, self.noop_max + 1)
        obs = None
        for _ in range(noops):
            obs, _, done, _ = self.env.step(self.noop_action)
            if done:
                obs = self.env.reset(**kwargs)
        return obs

    def step(self, action):
        return self.env.step(action)


class FireResetEnv(gym.Wrapper):
    """Press FIRE after reset, for envs that wait for it."""

    def __init__(self, env):
        super().__init__(env)
        meanings = env.unwrapped.get_action_meanings()
        assert meanings[1] == "FIRE" and len(meanings) >= 3

    def reset(self, **kwargs):
        self.env.reset(**kwargs)
        obs, _, done, _ = self.env.step(1)
        if done:
            self.env.reset(**kwargs)
        obs, _, done, _ = self.env.step(2)
        if done:
            self.env.reset(**kwargs)
        return obs

    def step(self, action):
        return self.env.step(action)


class EpisodicLifeEnv(gym.Wrapper):
    """Signal done on life loss, but only truly reset when the game ends."""

    def __init__(self, env):
        super().__init__(env)
        self.lives = 0
        self.was_real_done = True

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        self.was_real_done = done
        lives = self.env.unwrapped.ale.lives()
        if 0 < lives < self.lives:
            done = True
        self.lives = lives
        return obs, reward, done, info

    def reset(self, **kwargs):
        if self.was_real_done:
            obs = self.env.reset(**kwargs)
        else:
            obs, _, _, _ = self.env.step(0)
        self.lives = self.env.unwrapped.ale.lives()
        return obs


class MaxAndSkipEnv(gym.Wrapper):
    """Repeat each action `skip` times; observe the max of the last two frames."""

    def __init__(self, env, skip=4):
        super().__init__(env)
        self._obs_buffer = np.zeros((2,) + env.observation_space.shape, dtype=np.uint8)
        self._skip = skip

    def step(self, action):
        total_reward, done, info = 0.0, False, {}
        for i in range(self._skip):
            obs, reward, done, info = self.env.step(action)
            if i == self._skip - 2:
                self._obs_buffer[0] = obs
            if i == self._skip - 1:
                self._obs_buffer[1] = obs
            total_reward += reward
            if done:
                break
        return self._obs_buffer.max(axis=0), total_reward, done, info

    def reset(self, **kwargs):
        return self.env.reset(**kwargs)


class ClipRewardEnv(gym.RewardWrapper):
    def reward(self, reward):
        return float(np.sign(reward))


class WarpFrame(gym.ObservationWrapper):
    """Resize to width x height, optionally grayscale."""

    def __init__(self, env, width=84, height=84, grayscale=True):
        super().__init__(env)
        if cv2 is None:
            raise ImportError("WarpFrame requires opencv-python (cv2)")
        self.width, self.height, self.grayscale = width, height, grayscale
        channels = 1 if grayscale else 3
        self.observation_space = spaces.Box(
            low=0, high=255, shape=(height, width, channels), dtype=np.uint8
        )

    def observation(self, frame):
        if self.grayscale:
            frame = cv2.cvtColor(frame, cv2.COLOR_RGB2GRAY)
        frame = cv2.resize(
            frame, (self.width, self.height), interpolation=cv2.INTER_AREA
        )
        if self.grayscale:
            frame = np.expand_dims(frame, -1)
        return frame


class LazyFrames:
    """Shares frames between stacked observations to save replay memory."""

    def __init__(self, frames):
        self._frames = frames
        self._out = None

    def _force(self):
        if self._out is None:
            self._out = np.concatenate(self._frames, axis=-1)
            self._frames = None
        return self._out

    def __array__(self, dtype=None):
        out = self._force()
        return out.astype(dtype) if dtype is not None else out

    def __len__(self):
        return len(self._force())

    def __getitem__(self, i):
        return self._force()[i]


class FrameStack(gym.Wrapper):
    def __init__(self, env, k):
        super().__init__(env)
        self.k = k
        self.frames = collections.deque([], maxlen=k)
        shp = env.observation_space.shape
        self.observation_space = spaces.Box(
            low=0,
            high=255,
            shape=(shp[0], shp[1], shp[2] * k),
            dtype=env.observation_space.dtype,
        )

    def reset(self):
        obs = self.env.reset()
        for _ in range(self.k):
            self.frames.append(obs)
        return self._get_obs()

    def step(self, action):
        obs, reward, done, info = self.env.step(action)
        self.frames.append(obs)
        return self._get_obs(), reward, done, info

    def _get_obs(self):
        assert len(self.frames) == self.k
        return LazyFrames(list(self.frames))


class ScaledFloatFrame(gym.ObservationWrapper):
    def __init__(self, env):
        super().__init__(env)
        self.observation_space = spaces.Box(
            low=0, high=1, shape=env.observation_space.shape, dtype=np.float32
        )

    def observation(self, observation):
        return np.array(observation).astype(np.float32) / 255.0


class ImageToPyTorch(gym.ObservationWrapper):
    """HWC -> CHW."""

    def __init__(self, env):
        super().__init__(env)
        old = self.observation_space.shape
        self.observation_space = spaces.Box(
            low=0,
            high=255,
            shape=(old[-1], old[0], old[1]),
            dtype=np.uint8,
        )

    def observation(self, observation):
        return np.transpose(observation, axes=(2, 0, 1))


def make_atari(env_id):
    env = gym.make(env_id)
    assert "NoFrameskip" in env.spec.id
    env = NoopResetEnv(env, noop_max=30)
    env = MaxAndSkipEnv(env, skip=4)
    return env


def wrap_deepmind(env, episode_life=True, clip_rewards=True, frame_stack=False, scale=False):
    if episode_life:
        env = EpisodicLifeEnv(env)
    if "FIRE" in env.unwrapped.get_action_meanings():
        env = FireResetEnv(env)
    env = WarpFrame(env)
    if scale:
        env = ScaledFloatFrame(env)
    if clip_rewards:
        env = ClipRewardEnv(env)
    if frame_stack:
        env = FrameStack(env, 4)
    return env


def wrap_pytorch(env):
    return ImageToPyTorch(env)
