"""Environment-to-tensor adapter (capability parity with
torchbeast/core/environment.py).

Adapts any object speaking the classic gym API (`reset() -> obs`,
`step(a) -> (obs, reward, done, info)`) — which the synthetic envs also
implement — to the dict-of-tensors step protocol used across the framework:
every field carries leading [T=1, B=1] dims, episodes auto-reset on done
(the returned frame is already the new episode's first observation while
the bookkeeping fields describe the episode that just finished), and
episode_return / episode_step are tracked here so actors don't have to.
"""

import torch


class Environment:
    FIELDS = ("frame", "reward", "done", "episode_return", "episode_step",
              "last_action")

    def __init__(self, env):
        self.env = env
        self._ret = None
        self._steps = None

    @staticmethod
    def _wrap_frame(obs):
        t = obs if torch.is_tensor(obs) else torch.from_numpy(obs)
        return t.view(1, 1, *t.shape)

    def _packet(self, frame, reward, done, last_action):
        return {
            "frame": self._wrap_frame(frame),
            "reward": torch.tensor(reward, dtype=torch.float32).view(1, 1),
            "done": torch.tensor(done).view(1, 1),
            "episode_return": self._ret,
            "episode_step": self._steps,
            "last_action": last_action.view(1, 1),
        }

    def initial(self):
        self._ret = torch.zeros(1, 1)
        self._steps = torch.zeros(1, 1, dtype=torch.int32)
        return self._packet(
            self.env.reset(),
            reward=0.0,
            done=True,  # Marks an episode boundary for recurrent resets.
            last_action=torch.zeros(1, 1, dtype=torch.int64),
        )

    def step(self, action):
        obs, reward, done, _info = self.env.step(action.item())
        self._steps = self._steps + 1
        self._ret = self._ret + reward
        packet_ret, packet_steps = self._ret, self._steps
        if done:
            obs = self.env.reset()
            self._ret = torch.zeros(1, 1)
            self._steps = torch.zeros(1, 1, dtype=torch.int32)

        out = self._packet(obs, reward, done, action)
        # Report the (possibly just-finished) episode's bookkeeping.
        out["episode_return"] = packet_ret
        out["episode_step"] = packet_steps
        return out

    def close(self):
        if hasattr(self.env, "close"):
            self.env.close()
