"""Deep IMPALA ResNet (ref: torchbeast/polybeast_learner.py:134-266).

On GPU the 84x84 trunk runs on hand-written MFMA kernels: every 3x3 s1 p1
conv after the first goes through ops.functional.resnet_conv3x3 (bf16
channels_last activations end to end; conv_mfma.hip MODE 5/3 + wgrad),
with ReLU / max-pool / residual adds on bf16 channels_last ATen ops. The
first conv (CI = observation channels, e.g. 4) stays on ATen — its K-runs
are narrower than an MFMA A-fragment. Features are flattened in logical
NCHW order in both paths so checkpoints transfer between CPU and GPU.

Three feature sections with channel widths [16, 32, 32]; each section is a
3x3 s1 p1 conv, a 3x3 s2 p1 max-pool, and two residual blocks of
(ReLU, conv3x3, ReLU, conv3x3) with identity skip. For 84x84x4 inputs the
trunk output is 11*11*32 = 3872 features -> FC 256. The core input is the FC
output concatenated with the clipped reward (no one-hot action, matching the
reference deep net); optional single-layer LSTM with hidden size 256 and
done-masked state resets. Returns a (action, policy_logits, baseline) tuple
like the reference polybeast net.
"""

import os

import torch
from torch import nn
from torch.nn import functional as F


class _ResidualBlock(nn.Module):
    def __init__(self, channels):
        super().__init__()
        self.conv0 = nn.Conv2d(channels, channels, kernel_size=3, padding=1)
        self.conv1 = nn.Conv2d(channels, channels, kernel_size=3, padding=1)

    def forward(self, x):
        out = self.conv0(F.relu(x))
        out = self.conv1(F.relu(out))
        return out + x


class _Section(nn.Module):
    def __init__(self, in_channels, out_channels):
        super().__init__()
        self.conv = nn.Conv2d(in_channels, out_channels, kernel_size=3, padding=1)
        self.res0 = _ResidualBlock(out_channels)
        self.res1 = _ResidualBlock(out_channels)

    def forward(self, x):
        x = self.conv(x)
        x = F.max_pool2d(x, kernel_size=3, stride=2, padding=1)
        return self.res1(self.res0(x))


class ResNet(nn.Module):
    def __init__(self, observation_shape=(4, 84, 84), num_actions=6, use_lstm=False):
        super().__init__()
        self.observation_shape = observation_shape
        self.num_actions = num_actions
        self.use_lstm = use_lstm

        widths = [16, 32, 32]
        sections = []
        in_channels = observation_shape[0]
        for w in widths:
            sections.append(_Section(in_channels, w))
            in_channels = w
        self.feat_extract = nn.Sequential(*sections)

        with torch.no_grad():
            trunk_out = self.feat_extract(torch.zeros(1, *observation_shape)).numel()

        self.fc = nn.Linear(trunk_out, 256)

        # FC output ⊕ clipped reward.
        core_size = self.fc.out_features + 1
        if use_lstm:
            self.core = nn.LSTM(core_size, 256, num_layers=1)
            core_size = 256

        self.policy = nn.Linear(core_size, num_actions)
        self.baseline = nn.Linear(core_size, 1)

    def _features_mfma(self, x):
        """84x84 trunk on the MFMA conv kernels, bf16 channels_last."""
        from torchbeast_amd.ops import functional as tbf

        cl = torch.channels_last

        def res_block(blk, x):
            out = tbf.resnet_conv3x3(blk.conv0, F.relu(x))
            out = tbf.resnet_conv3x3(blk.conv1, F.relu(out))
            return out + x

        s1, s2, s3 = self.feat_extract
        # First conv: obs channels zero-padded to 8 on the same MFMA
        # template (wgrad-only backward; frames carry no grad).
        x = F.max_pool2d(tbf.resnet_first_conv(s1.conv, x), 3, 2, 1)
        x = x.contiguous(memory_format=cl)
        x = res_block(s1.res1, res_block(s1.res0, x))
        x = F.max_pool2d(tbf.resnet_conv3x3(s2.conv, x), 3, 2, 1)
        x = res_block(s2.res1, res_block(s2.res0, x))
        x = F.max_pool2d(tbf.resnet_conv3x3(s3.conv, x), 3, 2, 1)
        x = res_block(s3.res1, res_block(s3.res0, x))
        return x.float()

    def _features(self, x):
        if (x.is_cuda and x.shape[2:] == (84, 84) and x.shape[1] <= 8
                and os.environ.get("TBAMD_RESNET") != "aten"):
            from torchbeast_amd.ops import functional as tbf

            if tbf.resnet_conv3x3_supported(16, 42, 16):
                return self._features_mfma(x)
        return self.feat_extract(x)

    def initial_state(self, batch_size=1):
        if not self.use_lstm:
            return tuple()
        return tuple(
            torch.zeros(self.core.num_layers, batch_size, self.core.hidden_size)
            for _ in range(2)
        )

    def forward(self, inputs, core_state):
        frame, reward, done = inputs["frame"], inputs["reward"], inputs["done"]
        T, B = frame.shape[:2]
        x = torch.flatten(frame, 0, 1).float() / 255.0
        x = self._features(x)
        # reshape (not view): the MFMA path returns channels_last; logical
        # NCHW flatten order is preserved either way.
        x = F.relu(self.fc(F.relu(x).reshape(T * B, -1)))

        clipped_reward = torch.clamp(reward, -1, 1).view(T * B, 1)
        core_input = torch.cat([x, clipped_reward], dim=-1)

        if self.use_lstm:
            from torchbeast_amd.ops import functional as tbf

            core_input = core_input.view(T, B, -1)
            notdone = (~done).float()
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

        return (action, policy_logits, baseline), core_state


# polybeast-compatible alias (ref: polybeast_learner.py:134).
Net = ResNet
