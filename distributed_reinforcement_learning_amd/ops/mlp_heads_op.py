"""Fused policy+value MLP heads (K4) autograd wrapper.

Forward: one kernel runs both three-layer heads for a block of rows with
weights streamed through LDS (fused bias+ReLU, activations stashed bf16).
Backward: one kernel fuses the dgrad chains, ReLU masks and bias grads; the
six wgrad GEMMs ([256 x N x 256], MFMA-efficient) stay on hipBLASLt.
"""

from __future__ import annotations

from typing import Tuple

import torch

from distributed_reinforcement_learning_amd import ops as _ops


class _FusedMlpHeads(torch.autograd.Function):
    @staticmethod
    def forward(ctx, h: torch.Tensor,
                w1p, b1p, w2p, b2p, w3p, b3p,
                w1v, b1v, w2v, b2v, w3v, b3v):
        ext = _ops.require_ext()
        A = w3p.shape[0]
        weights = [w1p, w2p, w3p, w1v, w2v, w3v]
        biases = [b.float().contiguous() for b in (b1p, b2p, b3p,
                                                   b1v, b2v, b3v)]
        logits, value, stash = ext.mlp_heads_fwd(
            h.contiguous(), [w.contiguous() for w in weights], biases, A)
        ctx.save_for_backward(stash, *weights)
        ctx.A = A
        ctx.N = h.shape[0]
        return logits, value

    @staticmethod
    def backward(ctx, dlogits: torch.Tensor, dvalue: torch.Tensor):
        stash, w1p, w2p, w3p, w1v, w2v, w3v = ctx.saved_tensors
        ext = _ops.require_ext()
        N, A = ctx.N, ctx.A
        dev = stash.device
        if dlogits is None:
            dlogits = torch.zeros(N, A, dtype=torch.bfloat16, device=dev)
        if dvalue is None:
            dvalue = torch.zeros(N, dtype=torch.float32, device=dev)
        dlogits = dlogits.to(torch.bfloat16).contiguous()
        dvalue = dvalue.float().contiguous()
        # the dgrad kernel consumes PRE-TRANSPOSED weights (its A-operand
        # streams matrix rows); layer-3 policy is zero-padded to K=32
        wT3p = torch.nn.functional.pad(
            w3p.t().contiguous(), (0, 32 - A))
        wT = [w1p.t().contiguous(), w2p.t().contiguous(), wT3p,
              w1v.t().contiguous(), w2v.t().contiguous(),
              w3v.contiguous()]
        (dz1p, dz2p, dz1v, dz2v, dh, db1p, db2p, db3p, db1v, db2v,
         db3v) = ext.mlp_heads_bwd(dlogits, dvalue, stash, wT, A)
        soff = N * 256
        flat = stash.reshape(-1)
        a1p = flat[0:soff].reshape(N, 256)
        a2p = flat[soff:2 * soff].reshape(N, 256)
        a1v = flat[2 * soff:3 * soff].reshape(N, 256)
        a2v = flat[3 * soff:4 * soff].reshape(N, 256)
        hb = flat[4 * soff:5 * soff].reshape(N, 256)
        # wgrads: dW = dz^T @ a_prev (torch Linear convention W [out,in])
        dw1p = dz1p.t().mm(hb)
        dw2p = dz2p.t().mm(a1p)
        dw3p = dlogits.t().mm(a2p)
        dw1v = dz1v.t().mm(hb)
        dw2v = dz2v.t().mm(a1v)
        dw3v = dvalue.to(torch.bfloat16).unsqueeze(0).mm(a2v)
        bf = torch.bfloat16
        return (dh, dw1p, db1p.to(bf), dw2p, db2p.to(bf), dw3p,
                db3p.to(bf), dw1v, db1v.to(bf), dw2v, db2v.to(bf), dw3v,
                db3v.to(bf))


def fused_mlp_heads(h: torch.Tensor, policy_head, value_head
                    ) -> Tuple[torch.Tensor, torch.Tensor]:
    """h [N,256] f32 (grad ok); heads are MLPHead(256,[256,256],out) bf16.
    Returns (logits bf16 [N,A], value f32 [N])."""
    return _FusedMlpHeads.apply(
        h,
        policy_head.hidden[0].weight, policy_head.hidden[0].bias,
        policy_head.hidden[1].weight, policy_head.hidden[1].bias,
        policy_head.out.weight, policy_head.out.bias,
        value_head.hidden[0].weight, value_head.hidden[0].bias,
        value_head.hidden[1].weight, value_head.hidden[1].bias,
        value_head.out.weight, value_head.out.bias)


def heads_fusable(model, h: torch.Tensor) -> bool:
    ph, vh = model.policy_head, model.value_head
    return (h.is_cuda and h.shape[-1] == 256
            and ph.out.weight.dtype == torch.bfloat16
            and len(ph.hidden) == 2
            and all(l.out_features == 256 for l in ph.hidden)
            and all(l.out_features == 256 for l in vh.hidden)
            and ph.out.out_features <= 32 and vh.out.out_features == 1
            and _ops.available())
