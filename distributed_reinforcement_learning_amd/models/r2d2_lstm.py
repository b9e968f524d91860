"""R2D2 recurrent dueling-Q model (reference model/r2d2_lstm.py).

conv (single-channel POMDP frames) -> action embedding -> concat -> LSTM(64)
-> dense 128 ReLU -> Q = value(|A|) - mean(1) (r2d2_lstm.py:26-49).

Unlike IMPALA, R2D2 carries the LSTM state *through* the sequence, resetting
h/c to zero after any done step (r2d2_lstm.py:80-82,109-111). The conv/
embedding front end is still batched over B*L in one pass (it has no
sequential dependency); only the tiny hidden-64 cell recurrence loops over L.
"""

from __future__ import annotations

import torch
import torch.nn as nn
import torch.nn.functional as F

from distributed_reinforcement_learning_amd.models.blocks import (
    ActionEmbedding, AtariConvStack, LSTMCellTF, MLPHead,
)


class R2D2LstmQ(nn.Module):
    def __init__(self, input_shape, num_action: int, lstm_size: int = 64):
        super().__init__()
        h, w, c = input_shape
        assert (h, w) == (84, 84)
        self.num_action = num_action
        self.lstm_size = lstm_size
        self.conv = AtariConvStack(c)
        self.action_emb = ActionEmbedding(num_action)
        feat = self.conv.out_features + self.action_emb.out_features
        self.lstm = LSTMCellTF(feat, lstm_size)
        self.trunk = nn.Linear(lstm_size, 128)
        self.value_out = nn.Linear(128, num_action)
        self.mean_out = nn.Linear(128, 1)

    def _head(self, h: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.trunk(h.to(self.trunk.weight.dtype)))
        return self.value_out(x) - self.mean_out(x)

    def features(self, state: torch.Tensor,
                 prev_action: torch.Tensor) -> torch.Tensor:
        return torch.cat(
            [self.conv(state), self.action_emb(prev_action)], dim=1)

    def single_step(self, state: torch.Tensor, prev_action: torch.Tensor,
                    h: torch.Tensor, c: torch.Tensor):
        """state [N,84,84,C] normalized NHWC float. Returns (q, h', c')."""
        feat = self.features(state, prev_action)
        new_h, new_c = self.lstm(feat, h, c)
        return self._head(new_h), new_h, new_c

    def forward(self, state, prev_action, h, c):
        return self.single_step(state, prev_action, h, c)

    def unroll_sequence(self, seq_state: torch.Tensor,
                        seq_prev_action: torch.Tensor,
                        h0: torch.Tensor, c0: torch.Tensor,
                        seq_done: torch.Tensor):
        """Sequential unroll with done-masked state reset.

        seq_state [B,L,84,84,C], seq_prev_action [B,L], h0/c0 [B,H],
        seq_done [B,L] bool. Returns q_stack [B,L,A].
        After step i: (h, c) *= (~done_i) — reference r2d2_lstm.py:80-82.
        """
        B, L = seq_state.shape[:2]
        feat = self.features(
            seq_state.reshape(B * L, *seq_state.shape[2:]),
            seq_prev_action.reshape(B * L),
        ).reshape(B, L, -1)
        h, c = h0, c0
        qs = []
        for i in range(L):
            h, c = self.lstm(feat[:, i], h, c)
            qs.append(self._head(h))
            keep = (~seq_done[:, i]).to(h.dtype).unsqueeze(1)
            h = h * keep
            c = c * keep
        return torch.stack(qs, dim=1)
