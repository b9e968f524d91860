"""Fused LSTM-cell elementwise tail (K3 in SURVEY.md §2.5).

The cell is gates = [x,h] @ W + b (a GEMM — hipBLASLt via torch.matmul)
followed by the gate nonlinearities and state blend. The tail is 7 elementwise
ops over [N, 4H]/[N, H]; fusing them into one kernel (forward + backward)
removes ~10 kernel launches and 6 HBM round-trips per cell step.

Gate order [i, g, f, o], TF forget_bias added to f (models/blocks.LSTMCellTF).
"""

from __future__ import annotations

from typing import Tuple

import torch

from distributed_reinforcement_learning_amd import ops as _ops


class _FusedLstmTail(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gates: torch.Tensor, c_prev: torch.Tensor,
                forget_bias: float):
        ext = _ops.require_ext()
        # bf16 gates (straight from the addmm) are consumed natively —
        # no [N,4H] f32 cast on either side of the tail
        new_h, new_c, stash = ext.lstm_tail_fwd(gates.contiguous(),
                                                c_prev.contiguous(),
                                                forget_bias)
        ctx.save_for_backward(stash, c_prev, new_c)
        ctx.bf16_gates = gates.dtype == torch.bfloat16
        # no materialized grad_c: the batched unroll consumes only new_h,
        # so autograd would otherwise zero-fill a [N,H] every step
        ctx.set_materialize_grads(False)
        return new_h, new_c

    @staticmethod
    def backward(ctx, grad_h: torch.Tensor, grad_c: torch.Tensor):
        stash, c_prev, new_c = ctx.saved_tensors
        ext = _ops.require_ext()
        if grad_h is None:
            grad_h = torch.zeros(c_prev.shape, dtype=torch.float32,
                                 device=c_prev.device)
        if not (grad_h.dim() == 2 and grad_h.stride(1) == 1):
            grad_h = grad_h.contiguous()
        grad_gates, grad_c_prev = ext.lstm_tail_bwd(
            grad_h, None if grad_c is None else grad_c.contiguous(),
            stash, c_prev.contiguous(), new_c, ctx.bf16_gates)
        return grad_gates, grad_c_prev, None


def lstm_fused_step(gates: torch.Tensor, c_prev: torch.Tensor,
                    forget_bias: float = 1.0
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """gates [N,4H] (pre-activation, order i,g,f,o; f32 or bf16 on GPU),
    c_prev [N,H] -> (new_h, new_c) f32."""
    if gates.is_cuda:
        return _FusedLstmTail.apply(gates, c_prev, forget_bias)
    gates = gates.float()
    i, g, f, o = gates.chunk(4, dim=1)
    new_c = torch.sigmoid(f + forget_bias) * c_prev \
        + torch.sigmoid(i) * torch.tanh(g)
    new_h = torch.sigmoid(o) * torch.tanh(new_c)
    return new_h, new_c


class _LstmSeqTrain(torch.autograd.Function):
    """Whole done-masked LSTM recurrence (R2D2 trained window) as ONE
    kernel each way; dWh/dbias/x-projection grads are GEMM-shaped and stay
    outside (ops/hip/lstm_gates.hip seq-train pair)."""

    @staticmethod
    def forward(ctx, xg, wh, h0, c0, done, forget_bias):
        ext = _ops.require_ext()
        wh = wh.contiguous()
        done = done.contiguous()
        h_out, h_fin, c_fin, acts, c_prev, h_prev = ext.lstm_seq_train_fwd(
            xg.contiguous(), wh, h0.contiguous(),
            c0.contiguous(), done, forget_bias)
        ctx.save_for_backward(acts, c_prev, h_prev, wh, done)
        ctx.set_materialize_grads(False)
        return h_out, h_fin, c_fin

    @staticmethod
    def backward(ctx, dh_out, dh_fin, dc_fin):
        acts, c_prev, h_prev, wh, done = ctx.saved_tensors
        ext = _ops.require_ext()
        B, L, H = c_prev.shape
        if dh_out is None:
            dh_out = torch.zeros(B, L, H, dtype=torch.float32,
                                 device=c_prev.device)
        dxg, dh0, dc0 = ext.lstm_seq_train_bwd(
            dh_out.float().contiguous(),
            None if dh_fin is None else dh_fin.float().contiguous(),
            None if dc_fin is None else dc_fin.float().contiguous(),
            acts, c_prev, wh, done)
        # dWh = sum_t h_prev_t^T dgates_t — one GEMM over B*L rows
        dwh = h_prev.reshape(B * L, H).t().mm(dxg.reshape(B * L, 4 * H))
        return dxg, dwh, dh0, dc0, None, None


def lstm_seq_train(xg, wh, h0, c0, done, forget_bias: float = 1.0):
    """xg [B,L,4H] bf16 (x-projection + bias, grad ok), wh [H,4H] bf16
    (grad ok), h0/c0 [B,H] f32, done [B,L] bool ->
    (h_out [B,L,H] f32, h_fin, c_fin)."""
    return _LstmSeqTrain.apply(xg, wh, h0, c0, done, float(forget_bias))
