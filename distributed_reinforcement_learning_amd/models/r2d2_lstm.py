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
    ActionEmbedding, AtariConvStack, LSTMCellTF,
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
        # value(|A|) and mean(1) branches share ONE [128, A+1] GEMM
        # (columns [:A] = value, column A = mean) — identical math to the
        # reference's two dense layers (r2d2_lstm.py:46-49) at half the
        # output-GEMM launches; the dueling subtract slices the result
        self.out = nn.Linear(128, num_action + 1)

    def _head(self, h: torch.Tensor) -> torch.Tensor:
        x = F.relu(self.trunk(h.to(self.trunk.weight.dtype)))
        y = self.out(x)
        return y[..., :self.num_action] - y[..., self.num_action:]

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

        When no gradient is required on GPU (burn-in recompute, target-net
        unroll, TD scoring), the whole recurrence runs as ONE kernel
        (ops/hip drla_lstm_seq_fwd): the x-projection is a single batched
        GEMM and only the tiny h @ Wh chain loops inside the kernel.
        """
        B, L = seq_state.shape[:2]
        feat = self.features(
            seq_state.reshape(B * L, *seq_state.shape[2:]),
            seq_prev_action.reshape(B * L),
        ).reshape(B, L, -1)

        kern_ok = (feat.is_cuda and self.lstm.weight.dtype == torch.bfloat16
                   and 4 * self.lstm_size <= 1024)
        if kern_ok and not torch.is_grad_enabled():
            from distributed_reinforcement_learning_amd import ops as _o
            ext = _o.require_ext()
            F = feat.shape[-1]
            w = self.lstm.weight
            xg = feat.reshape(B * L, F) @ w[:F] + self.lstm.bias
            h_all, _, _ = ext.lstm_seq_fwd(
                xg.reshape(B, L, -1).contiguous(),
                w[F:].contiguous(), h0.float().contiguous(),
                c0.float().contiguous(), seq_done.contiguous(),
                self.lstm.forget_bias)
            q = self._head(h_all.reshape(B * L, -1))
            return q.reshape(B, L, -1)
        if kern_ok:
            # trained window: grad-carrying one-kernel recurrence
            # (ops/lstm_op.lstm_seq_train); x-projection, dWh and bias
            # grads are GEMM-shaped and ride on autograd/hipBLASLt
            from distributed_reinforcement_learning_amd.ops.lstm_op import (
                lstm_seq_train,
            )
            F = feat.shape[-1]
            w = self.lstm.weight
            xg = torch.addmm(self.lstm.bias, feat.reshape(B * L, F), w[:F])
            h_all, _, _ = lstm_seq_train(
                xg.reshape(B, L, -1), w[F:], h0.float(), c0.float(),
                seq_done, self.lstm.forget_bias)
            import os as _os
            # fused train-head measured NET SLOWER than the torch chain
            # (A/B same box: 22.2k -> 20.0k seq/s at 15/7 — the fused
            # backward's serial per-row phases cost more than the ~13
            # launches it saves); opt-in for round-3 tuning
            if (self.out.weight.dtype == torch.bfloat16
                    and self.trunk.weight.dtype == torch.bfloat16
                    and _os.environ.get("DRLA_FUSED_DHEAD") == "1"):
                from distributed_reinforcement_learning_amd.ops import (
                    r2d2_op,
                )
                fused_dueling_head_train = r2d2_op.fused_dueling_head_train
                q = fused_dueling_head_train(
                    h_all.reshape(B * L, -1), self.trunk, self.out)
                return q.reshape(B, L, -1)
            q = self._head(h_all.reshape(B * L, -1))
            return q.reshape(B, L, -1)

        h, c = h0, c0
        qs = []
        for i in range(L):
            h, c = self.lstm(feat[:, i], h, c)
            qs.append(self._head(h))
            keep = (~seq_done[:, i]).to(h.dtype).unsqueeze(1)
            h = h * keep
            c = c * keep
        return torch.stack(qs, dim=1)

    @torch.no_grad()
    def q_window(self, seq_state: torch.Tensor,
                 seq_prev_action: torch.Tensor, h0: torch.Tensor,
                 c0: torch.Tensor, seq_done: torch.Tensor, burn: int):
        """No-grad post-burn-in Q window in ONE front-end pass: conv/
        embed/x-projection over the FULL [B,L] sequence, one seq-recurrence
        kernel, head only on the trained window. Replaces the
        burn_in_states + unroll_sequence pair for the target net and the
        scoring path (same math, half the launches)."""
        B, L = seq_state.shape[:2]
        feat = self.features(
            seq_state.reshape(B * L, *seq_state.shape[2:]),
            seq_prev_action.reshape(B * L)).reshape(B, L, -1)
        if (feat.is_cuda and self.lstm.weight.dtype == torch.bfloat16
                and 4 * self.lstm_size <= 1024):
            from distributed_reinforcement_learning_amd import ops as _o
            ext = _o.require_ext()
            F_ = feat.shape[-1]
            w = self.lstm.weight
            xg = feat.reshape(B * L, F_) @ w[:F_] + self.lstm.bias
            h_all, _, _ = ext.lstm_seq_fwd(
                xg.reshape(B, L, -1).contiguous(), w[F_:].contiguous(),
                h0.float().contiguous(), c0.float().contiguous(),
                seq_done.contiguous(), self.lstm.forget_bias)
            if (self.out.weight.dtype == torch.bfloat16
                    and self.trunk.weight.dtype == torch.bfloat16):
                # whole dueling head in one launch (window slice resolved
                # by index math in-kernel; replaces cast/addmm/relu/addmm/
                # slice-sub)
                return ext.dueling_head_fwd(
                    h_all, self.trunk.weight.contiguous(),
                    self.trunk.bias.contiguous(),
                    self.out.weight.contiguous(),
                    self.out.bias.contiguous(), burn)
            # the head's dtype cast absorbs the window slice's strides
            return self._head(h_all[:, burn:])
        h, c = h0, c0
        qs = []
        for i in range(L):
            h, c = self.lstm(feat[:, i], h, c)
            if i >= burn:
                qs.append(self._head(h))
            keep = (~seq_done[:, i]).to(h.dtype).unsqueeze(1)
            h = h * keep
            c = c * keep
        return torch.stack(qs, dim=1)

    @torch.no_grad()
    def burn_in_states(self, seq_state: torch.Tensor,
                       seq_prev_action: torch.Tensor, h0: torch.Tensor,
                       c0: torch.Tensor, seq_done: torch.Tensor):
        """Hidden-state recompute over the burn-in window (no gradient, no
        Q heads) — BASELINE's "burn_in hidden-state recompute" hot-path
        item. Returns (h, c) to start the trained window from. One fused
        kernel on GPU (drla_lstm_seq_fwd), torch loop on CPU."""
        B, L = seq_state.shape[:2]
        feat = self.features(
            seq_state.reshape(B * L, *seq_state.shape[2:]),
            seq_prev_action.reshape(B * L)).reshape(B, L, -1)
        if (feat.is_cuda and self.lstm.weight.dtype == torch.bfloat16
                and 4 * self.lstm_size <= 1024):
            from distributed_reinforcement_learning_amd import ops as _o
            ext = _o.require_ext()
            F = feat.shape[-1]
            w = self.lstm.weight
            xg = feat.reshape(B * L, F) @ w[:F] + self.lstm.bias
            _, h_fin, c_fin = ext.lstm_seq_fwd(
                xg.reshape(B, L, -1).contiguous(), w[F:].contiguous(),
                h0.float().contiguous(), c0.float().contiguous(),
                seq_done.contiguous(), self.lstm.forget_bias)
            return h_fin, c_fin
        h, c = h0, c0
        for i in range(L):
            h, c = self.lstm(feat[:, i], h, c)
            keep = (~seq_done[:, i]).to(h.dtype).unsqueeze(1)
            h = h * keep
            c = c * keep
        return h, c
