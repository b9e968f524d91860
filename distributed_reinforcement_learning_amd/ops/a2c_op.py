"""Fused A2C loss pipeline (K6) autograd wrapper.

One kernel each way covers the A3C learner's whole post-network math
(reference optimizer/a2c.py:3-26 + agent/a3c.py:39-51 clip/discount glue):
softmax, 1-step-TD advantage, the three losses and the combined total
forward; closed-form dlogits/dvalue backward. CPU golden lives in
algorithms/a2c.py (the parity tests compare against it).
"""

from __future__ import annotations

from typing import Tuple

import torch

from distributed_reinforcement_learning_amd import ops as _ops

_CLIP_MODE = {"abs_one": 0, "soft_asymmetric": 1, "none": 2}


class _FusedA2cLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, value, next_value, actions, rewards, done,
                gamma, clip_mode, c_bl, c_ent):
        ext = _ops.require_ext()
        actions = actions.contiguous()
        losses, p_stash, adv_st = ext.a2c_loss_fwd(
            logits.contiguous(), value.float().contiguous(),
            next_value.float().contiguous(), actions,
            rewards.contiguous(), done.contiguous(), gamma, clip_mode,
            c_bl, c_ent)
        ctx.save_for_backward(p_stash, adv_st, actions)
        ctx.coefs = (c_bl, c_ent)
        ctx.want_bf16 = logits.dtype == torch.bfloat16
        ctx.set_materialize_grads(False)
        return losses[0], losses[1], losses[2], losses[3]

    @staticmethod
    def backward(ctx, g_pi, g_base, g_ent, g_total):
        p_stash, adv_st, actions = ctx.saved_tensors
        ext = _ops.require_ext()
        c_bl, c_ent = ctx.coefs
        none3 = g_pi is None and g_base is None and g_ent is None
        if g_total is not None and none3:
            grad3 = g_total.reshape(1).float().contiguous()
            from_total = True
        else:
            dev = p_stash.device
            z = torch.zeros((), dtype=torch.float32, device=dev)
            grad3 = torch.stack([
                g_pi if g_pi is not None else z,
                g_base if g_base is not None else z,
                g_ent if g_ent is not None else z,
            ]).float()
            if g_total is not None:
                grad3 = grad3 + g_total.float() * torch.tensor(
                    [1.0, c_bl, c_ent], device=dev)
            grad3 = grad3.contiguous()
            from_total = False
        dlogits, dvalue = ext.a2c_loss_bwd(
            p_stash, adv_st, actions, grad3, from_total, c_bl, c_ent,
            ctx.want_bf16)
        return (dlogits, dvalue) + (None,) * 8


def fused_a2c_loss(logits: torch.Tensor, value: torch.Tensor,
                   next_value: torch.Tensor, actions: torch.Tensor,
                   rewards: torch.Tensor, done: torch.Tensor, gamma: float,
                   reward_clipping: str, c_bl: float, c_ent: float
                   ) -> Tuple[torch.Tensor, ...]:
    """logits [N,A] (bf16/f32, grad ok), value [N] (grad ok), next_value
    [N] (detached in-kernel), actions [N] int, rewards [N] RAW, done [N]
    bool -> (pi_loss, baseline_loss, entropy, total)."""
    return _FusedA2cLoss.apply(
        logits, value, next_value.detach(), actions.to(torch.int32),
        rewards, done, float(gamma), _CLIP_MODE[reward_clipping],
        float(c_bl), float(c_ent))
