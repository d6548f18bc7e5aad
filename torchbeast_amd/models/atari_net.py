"""Shallow IMPALA Atari network (ref: torchbeast/monobeast.py:545-635).

Architecture (identical hyperparameters to the reference):
  conv 8x8 s4 C->32 + ReLU -> conv 4x4 s2 32->64 + ReLU -> conv 3x3 s1 64->64
  + ReLU -> FC 3136->512 + ReLU; core input = fc ⊕ one_hot(last_action)
  ⊕ clamp(reward, -1, 1); optional 2-layer LSTM with hidden == core input
  size, with per-timestep done-masked state resets; policy and baseline
  heads; multinomial sampling in train mode, argmax in eval.

MI355X path: when the input lives on a ROCm device and the HIP extension is
available, the conv trunk, the LSTM unroll and the heads dispatch to fused
CDNA4 kernels (see torchbeast_amd/ops/). The eager path below is the
CPU/oracle implementation.
"""

import torch
from torch import nn
from torch.nn import functional as F


class AtariNet(nn.Module):
    def __init__(self, observation_shape, num_actions, use_lstm=False,
                 use_last_action=True):
        super().__init__()
        self.observation_shape = observation_shape
        self.num_actions = num_actions
        self.use_lstm = use_lstm
        # The polybeast env plane doesn't transport last_action (the deep
        # reference net doesn't consume it either, ref:
        # polybeast_learner.py:234-235); monobeast's AtariNet does.
        self.use_last_action = use_last_action

        in_channels = observation_shape[0]
        self.conv1 = nn.Conv2d(in_channels, 32, kernel_size=8, stride=4)
        self.conv2 = nn.Conv2d(32, 64, kernel_size=4, stride=2)
        self.conv3 = nn.Conv2d(64, 64, kernel_size=3, stride=1)

        # 84x84 -> 20x20 -> 9x9 -> 7x7.
        conv_out = self._conv_out_elems(observation_shape)
        self.fc = nn.Linear(conv_out, 512)

        # FC output ⊕ clipped reward ⊕ (optional) one-hot last action.
        core_size = self.fc.out_features + 1 + (num_actions if use_last_action else 0)
        if use_lstm:
            self.core = nn.LSTM(core_size, core_size, num_layers=2)

        self.policy = nn.Linear(core_size, num_actions)
        self.baseline = nn.Linear(core_size, 1)

    def _conv_out_elems(self, shape):
        with torch.no_grad():
            x = torch.zeros(1, *shape)
            x = self.conv3(self.conv2(self.conv1(x)))
        return x.numel()

    def initial_state(self, batch_size):
        if not self.use_lstm:
            return tuple()
        return tuple(
            torch.zeros(self.core.num_layers, batch_size, self.core.hidden_size)
            for _ in range(2)
        )

    def forward(self, inputs, core_state=()):
        from torchbeast_amd.ops import functional as tbf

        frame = inputs["frame"]  # [T, B, C, H, W], uint8.
        T, B = frame.shape[:2]
        flat_frames = torch.flatten(frame, 0, 1)
        # GPU: fused u8 conv trunk (one kernel); CPU/oversized: eager chain.
        x = tbf.atari_trunk(flat_frames, self.conv1, self.conv2, self.conv3)
        if x is None:
            x = flat_frames.float() / 255.0
            x = F.relu(self.conv1(x))
            x = F.relu(self.conv2(x))
            x = F.relu(self.conv3(x))
            x = x.view(T * B, -1)
        x = F.relu(self.fc(x))

        clipped_reward = torch.clamp(inputs["reward"], -1, 1).view(T * B, 1)
        if self.use_last_action:
            last_action = torch.flatten(inputs["last_action"], 0, 1)
            one_hot_action = F.one_hot(last_action, self.num_actions).float()
            core_input = torch.cat([x, clipped_reward, one_hot_action], dim=-1)
        else:
            core_input = torch.cat([x, clipped_reward], dim=-1)

        if self.use_lstm:
            from torchbeast_amd.ops import functional as tbf

            core_input = core_input.view(T, B, -1)
            notdone = (~inputs["done"]).float()
            # Done-masked unroll: one fused HIP kernel on GPU, a Python
            # step loop on CPU (torchbeast_amd/ops/functional.py).
            core_output_seq, core_state = tbf.lstm_unroll(
                self.core, core_input, notdone, core_state
            )
            core_output = torch.flatten(core_output_seq, 0, 1)
        else:
            core_output = core_input

        policy_logits = self.policy(core_output)
        baseline = self.baseline(core_output)

        from torchbeast_amd.ops import functional as tbf

        action = tbf.policy_sample(policy_logits, greedy=not self.training)

        policy_logits = policy_logits.view(T, B, self.num_actions)
        baseline = baseline.view(T, B)
        action = action.view(T, B)

        return (
            dict(policy_logits=policy_logits, baseline=baseline, action=action),
            core_state,
        )


# The reference exports the shallow net under both names (monobeast.py:635).
Net = AtariNet
