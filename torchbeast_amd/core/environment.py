"""Environment-to-tensor adapter (ref: torchbeast/core/environment.py).

Wraps any object with `reset() -> obs` and `step(a) -> (obs, reward, done,
info)` (gym classic API, which the synthetic envs also speak) into the
dict-of-tensors protocol used throughout the framework: every field carries
leading [T=1, B=1] dims, episodes auto-reset on done, and `episode_return` /
`episode_step` are tracked here so actors don't have to.
"""

import torch


def _frame_to_tensor(frame):
    t = torch.from_numpy(frame) if not torch.is_tensor(frame) else frame
    return t.view(1, 1, *t.shape)


class Environment:
    def __init__(self, env):
        self.env = env
        self.episode_return = None
        self.episode_step = None

    def initial(self):
        initial_frame = _frame_to_tensor(self.env.reset())
        self.episode_return = torch.zeros(1, 1)
        self.episode_step = torch.zeros(1, 1, dtype=torch.int32)
        return dict(
            frame=initial_frame,
            reward=torch.zeros(1, 1),
            done=torch.ones(1, 1, dtype=torch.bool),
            episode_return=self.episode_return,
            episode_step=self.episode_step,
            last_action=torch.zeros(1, 1, dtype=torch.int64),
        )

    def step(self, action):
        frame, reward, done, _ = self.env.step(action.item())
        self.episode_step += 1
        self.episode_return += reward
        episode_step = self.episode_step
        episode_return = self.episode_return
        if done:
            frame = self.env.reset()
            self.episode_return = torch.zeros(1, 1)
            self.episode_step = torch.zeros(1, 1, dtype=torch.int32)

        return dict(
            frame=_frame_to_tensor(frame),
            reward=torch.tensor(reward, dtype=torch.float32).view(1, 1),
            done=torch.tensor(done).view(1, 1),
            episode_return=episode_return,
            episode_step=episode_step,
            last_action=action.view(1, 1),
        )

    def close(self):
        if hasattr(self.env, "close"):
            self.env.close()
