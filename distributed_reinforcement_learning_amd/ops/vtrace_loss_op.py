"""Fused IMPALA loss pipeline (softmax + V-trace + three losses) with
closed-form backward — ops/hip/vtrace_loss.hip.

Autograd contract: returns the three loss scalars; when the caller backwards
``pi + c_b*baseline + c_e*entropy`` the upstream grads (1, c_b, c_e) arrive
as grad_outputs and flow straight into the backward kernel.
"""

from __future__ import annotations

from typing import Tuple

import torch

from distributed_reinforcement_learning_amd import ops as _ops


class _FusedVtraceLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, value: torch.Tensor,
                mu: torch.Tensor, actions: torch.Tensor,
                rewards: torch.Tensor, discounts: torch.Tensor):
        ext = _ops.require_ext()
        losses, p_stash, vs_stash, adv_stash = ext.vtrace_loss_fwd(
            logits.contiguous(), value.contiguous(), mu.contiguous(),
            actions.contiguous(), rewards.contiguous(),
            discounts.contiguous())
        ctx.save_for_backward(p_stash, vs_stash, adv_stash,
                              value.contiguous(), actions.contiguous())
        ctx.want_bf16 = logits.dtype == torch.bfloat16
        return losses[0], losses[1], losses[2]

    @staticmethod
    def backward(ctx, g_pi, g_base, g_ent):
        p_stash, vs_stash, adv_stash, value, actions = ctx.saved_tensors
        ext = _ops.require_ext()
        grad3 = torch.stack([g_pi, g_base, g_ent]).float().contiguous()
        dlogits, dvalue = ext.vtrace_loss_bwd(
            p_stash, vs_stash, adv_stash, value, actions, grad3,
            ctx.want_bf16)
        return dlogits, dvalue, None, None, None, None


def fused_vtrace_loss(logits: torch.Tensor, value: torch.Tensor,
                      mu: torch.Tensor, actions: torch.Tensor,
                      rewards: torch.Tensor, discounts: torch.Tensor
                      ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor]:
    """logits [B,T,A] (bf16/f32, requires_grad ok), value [B,T] f32
    (requires_grad ok), mu [B,T,A] f32, actions [B,T] int32,
    rewards/discounts [B,T] f32 -> (pi_loss, baseline_loss, entropy)."""
    return _FusedVtraceLoss.apply(logits, value, mu,
                                  actions.to(torch.int32), rewards,
                                  discounts)
