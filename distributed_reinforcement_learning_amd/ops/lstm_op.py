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
        return new_h, new_c

    @staticmethod
    def backward(ctx, grad_h: torch.Tensor, grad_c: torch.Tensor):
        stash, c_prev, new_c = ctx.saved_tensors
        ext = _ops.require_ext()
        if not (grad_h.dim() == 2 and grad_h.stride(1) == 1):
            grad_h = grad_h.contiguous()
        grad_gates, grad_c_prev = ext.lstm_tail_bwd(
            grad_h, grad_c.contiguous(), stash,
            c_prev.contiguous(), new_c, ctx.bf16_gates)
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
