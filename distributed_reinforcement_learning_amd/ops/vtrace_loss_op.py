"""Fused IMPALA loss pipeline (reward clip + discounts + softmax + V-trace
+ three losses + combined total) with closed-form backward —
ops/hip/vtrace_loss.hip.

Autograd contract: returns (pi, baseline, entropy, total) where total =
pi + c_bl*baseline + c_ent*entropy is accumulated IN-KERNEL. The standard
``total.backward()`` path seeds the backward kernel with the 1-element
upstream grad directly (from_total mode) — no stack/scale kernels; calling
backward through the individual losses still works (generic grad3 path).
"""

from __future__ import annotations

from typing import Tuple

import torch

from distributed_reinforcement_learning_amd import ops as _ops

_CLIP_MODE = {"abs_one": 0, "soft_asymmetric": 1, "none": 2}


class _FusedVtraceLoss(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, value: torch.Tensor,
                mu: torch.Tensor, actions: torch.Tensor,
                rewards: torch.Tensor, done: torch.Tensor, gamma: float,
                clip_mode: int, c_bl: float, c_ent: float):
        ext = _ops.require_ext()
        losses, p_stash, vs_stash, adv_stash = ext.vtrace_loss_fwd(
            logits.contiguous(), value.contiguous(), mu.contiguous(),
            actions.contiguous(), rewards.contiguous(), done.contiguous(),
            gamma, clip_mode, c_bl, c_ent)
        ctx.save_for_backward(p_stash, vs_stash, adv_stash,
                              value.contiguous(), actions.contiguous())
        ctx.want_bf16 = logits.dtype == torch.bfloat16
        ctx.coefs = (c_bl, c_ent)
        ctx.set_materialize_grads(False)
        return losses[0], losses[1], losses[2], losses[3]

    @staticmethod
    def backward(ctx, g_pi, g_base, g_ent, g_total):
        p_stash, vs_stash, adv_stash, value, actions = ctx.saved_tensors
        ext = _ops.require_ext()
        c_bl, c_ent = ctx.coefs
        none3 = g_pi is None and g_base is None and g_ent is None
        if g_total is not None and none3:
            # total.backward(): 1-element seed straight into the kernel
            grad3 = g_total.reshape(1)
            from_total = True
        else:
            dev = p_stash.device
            z = torch.zeros((), dtype=torch.float32, device=dev)
            grad3 = torch.stack([
                g_pi if g_pi is not None else z,
                g_base if g_base is not None else z,
                g_ent if g_ent is not None else z,
            ]).float().contiguous()
            if g_total is not None:
                grad3 = grad3 + g_total.float() * torch.tensor(
                    [1.0, c_bl, c_ent], device=dev)
            from_total = False
        dlogits, dvalue = ext.vtrace_loss_bwd(
            p_stash, vs_stash, adv_stash, value, actions,
            grad3.float().contiguous(), from_total, c_bl, c_ent,
            ctx.want_bf16)
        return (dlogits, dvalue) + (None,) * 8


def fused_vtrace_loss(logits: torch.Tensor, value: torch.Tensor,
                      mu: torch.Tensor, actions: torch.Tensor,
                      rewards: torch.Tensor, done: torch.Tensor,
                      gamma: float, reward_clipping: str, c_bl: float,
                      c_ent: float
                      ) -> Tuple[torch.Tensor, torch.Tensor, torch.Tensor,
                                 torch.Tensor]:
    """logits [B,T,A] (bf16/f32, grad ok), value [B,T] f32 (grad ok),
    mu [B,T,A] f32, actions [B,T] int, rewards [B,T] f32 RAW, done [B,T]
    bool -> (pi_loss, baseline_loss, entropy, total)."""
    return _FusedVtraceLoss.apply(logits, value, mu,
                                  actions.to(torch.int32), rewards, done,
                                  float(gamma), _CLIP_MODE[reward_clipping],
                                  float(c_bl), float(c_ent))
