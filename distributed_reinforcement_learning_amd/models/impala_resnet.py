"""IMPALA 'deep' ResNet actor-critic (Espeholt et al. 2018, the large
architecture) — BASELINE config #5: "IMPALA ResNet-large 84x84, 8xMI355X DP
all-reduce (xGMI scaling curve)".

Three sections of [16, 32, 32] channels, each = conv3x3 + maxpool/2 + two
residual blocks, then ReLU -> flatten -> fc256, feeding the same action
embedding + LSTM + heads as the shallow model. ~10x the conv FLOPs of the
Atari stack — the config that stresses the gradient all-reduce.

Runs channels_last on MIOpen (library conv path; the hand-written MFMA
kernels cover the flagship Atari stack — see ops/hip/conv.hip).
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from distributed_reinforcement_learning_amd.models.blocks import (
    ActionEmbedding, LSTMCellTF, MLPHead,
)


class _ResBlock(nn.Module):
    def __init__(self, ch: int):
        super().__init__()
        self.c1 = nn.Conv2d(ch, ch, 3, padding=1)
        self.c2 = nn.Conv2d(ch, ch, 3, padding=1)

    def forward(self, x):
        y = self.c1(F.relu(x))
        y = self.c2(F.relu(y))
        return x + y


class _Section(nn.Module):
    def __init__(self, cin: int, cout: int):
        super().__init__()
        self.conv = nn.Conv2d(cin, cout, 3, padding=1)
        self.r1 = _ResBlock(cout)
        self.r2 = _ResBlock(cout)

    def forward(self, x):
        x = self.conv(x)
        x = F.max_pool2d(x, 3, stride=2, padding=1)
        return self.r2(self.r1(x))


class ImpalaResNetTorso(nn.Module):
    def __init__(self, in_channels: int, channels=(16, 32, 32)):
        super().__init__()
        secs = []
        prev = in_channels
        for ch in channels:
            secs.append(_Section(prev, ch))
            prev = ch
        self.sections = nn.ModuleList(secs)
        # 84 -> 42 -> 21 -> 11 with pool padding 1
        self.out_features = channels[-1] * 11 * 11
        self.fc = nn.Linear(self.out_features, 256)

    def forward(self, x_nhwc: torch.Tensor) -> torch.Tensor:
        x = x_nhwc.to(self.fc.weight.dtype).permute(0, 3, 1, 2)
        if x.device.type == "cuda":
            x = x.contiguous(memory_format=torch.channels_last)
        else:
            x = x.contiguous()
        for s in self.sections:
            x = s(x)
        x = F.relu(x)
        x = x.permute(0, 2, 3, 1).flatten(1)  # NHWC flatten, as blocks.py
        return F.relu(self.fc(x))


class ImpalaResNetActorCritic(nn.Module):
    """Same interface as ImpalaActorCritic (single_step / unroll /
    unroll_logits)."""

    def __init__(self, input_shape, num_action: int,
                 lstm_hidden_size: int = 256):
        super().__init__()
        h, w, c = input_shape
        assert (h, w) == (84, 84)
        self.num_action = num_action
        self.lstm_hidden_size = lstm_hidden_size
        self.conv = ImpalaResNetTorso(c)
        self.action_emb = ActionEmbedding(num_action)
        feat = 256 + self.action_emb.out_features
        self.lstm = LSTMCellTF(feat, lstm_hidden_size)
        self.policy_head = MLPHead(lstm_hidden_size, [256, 256], num_action,
                                   "softmax")
        self.value_head = MLPHead(lstm_hidden_size, [256, 256], 1, None)

    def _features(self, state, prev_action):
        s = state
        if s.dtype == torch.uint8:
            # ResNet torso runs the library conv path: normalize here
            from distributed_reinforcement_learning_amd.ops import (
                normalize_frames,
            )
            s = normalize_frames(s, out_dtype=self.fcdtype())
        return torch.cat([self.conv(s), self.action_emb(prev_action)], dim=1)

    def fcdtype(self):
        return self.conv.fc.weight.dtype

    def single_step(self, state, prev_action, h, c):
        feat = self._features(state, prev_action)
        new_h, new_c = self.lstm(feat, h, c)
        policy = self.policy_head(new_h)
        value = self.value_head(new_h).squeeze(-1)
        return policy, value, new_h, new_c

    forward = single_step

    def _unroll_features(self, traj_state, traj_prev_action, traj_h, traj_c):
        B, T = traj_state.shape[:2]
        feat = self._features(
            traj_state.reshape(B * T, *traj_state.shape[2:]),
            traj_prev_action.reshape(B * T))
        new_h, _ = self.lstm(feat, traj_h.reshape(B * T, -1),
                             traj_c.reshape(B * T, -1))
        return new_h, B, T

    def unroll(self, traj_state, traj_prev_action, traj_h, traj_c):
        h, B, T = self._unroll_features(traj_state, traj_prev_action,
                                        traj_h, traj_c)
        return (self.policy_head(h).reshape(B, T, -1),
                self.value_head(h).squeeze(-1).reshape(B, T))

    def unroll_logits(self, traj_state, traj_prev_action, traj_h, traj_c):
        h, B, T = self._unroll_features(traj_state, traj_prev_action,
                                        traj_h, traj_c)
        return (self.policy_head.logits(h).reshape(B, T, -1),
                self.value_head(h).squeeze(-1).reshape(B, T))
