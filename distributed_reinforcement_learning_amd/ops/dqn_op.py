"""Fused double-DQN target + IS-weighted TD loss (K8) autograd wrapper."""

from __future__ import annotations

from typing import Tuple

import torch

from distributed_reinforcement_learning_amd import ops as _ops


class _FusedDqnLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, main_q: torch.Tensor, next_main_q: torch.Tensor,
                next_target_q: torch.Tensor, actions: torch.Tensor,
                rewards: torch.Tensor, discounts: torch.Tensor,
                weights: torch.Tensor):
        ext = _ops.require_ext()
        loss, td = ext.dqn_loss_fwd(
            main_q.contiguous(), next_main_q.float().contiguous(),
            next_target_q.float().contiguous(), actions.contiguous(),
            rewards.contiguous(), discounts.contiguous(),
            weights.contiguous())
        ctx.save_for_backward(td, actions, weights)
        ctx.A = main_q.shape[1]
        ctx.want_bf16 = main_q.dtype == torch.bfloat16
        return loss[0], td

    @staticmethod
    def backward(ctx, g_loss, g_td):
        td, actions, weights = ctx.saved_tensors
        ext = _ops.require_ext()
        dmq = ext.dqn_loss_bwd(td, actions, weights,
                               g_loss.reshape(1).float().contiguous(),
                               ctx.A, ctx.want_bf16)
        return dmq, None, None, None, None, None, None


def fused_dqn_loss(main_q: torch.Tensor, next_main_q: torch.Tensor,
                   next_target_q: torch.Tensor, actions: torch.Tensor,
                   rewards: torch.Tensor, discounts: torch.Tensor,
                   weights: torch.Tensor) -> Tuple[torch.Tensor,
                                                   torch.Tensor]:
    """Returns (loss scalar with grad to main_q, signed td [B] no-grad)."""
    return _FusedDqnLoss.apply(main_q, next_main_q, next_target_q,
                               actions.to(torch.int32), rewards, discounts,
                               weights)
