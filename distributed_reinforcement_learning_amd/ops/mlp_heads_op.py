"""Fused policy+value MLP heads (K4) autograd wrapper.

Forward: one kernel runs both three-layer heads for a block of rows with
weights streamed through LDS (fused bias+ReLU, activations stashed bf16).
Backward: one kernel fuses the dgrad chains, ReLU masks and f32 bias-grad
partials; a second MFMA kernel computes all six wgrads (dW = dz^T @ act)
plus the bf16 bias-grad conversion — so the whole bwd is 3 launches
(W^T pack, dgrad chain, wgrad) instead of ~20 torch ops.
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
        # biases are consumed bf16 in-kernel (no per-step f32 casts)
        biases = [b.contiguous() for b in (b1p, b2p, b3p, b1v, b2v, b3v)]
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
        # streams matrix rows); ONE pack kernel builds all of W^T + the
        # layer-3 zero-pad to K=32 (was 5 .t().contiguous() + F.pad)
        packed = ext.mlp_heads_pack_wt(
            [w1p, w2p, w3p, w1v, w2v, w3v], A)
        wT = [packed.narrow(0, 0, 65536).view(256, 256),
              packed.narrow(0, 65536, 65536).view(256, 256),
              packed.narrow(0, 131072, 8192).view(256, 32),
              packed.narrow(0, 139264, 65536).view(256, 256),
              packed.narrow(0, 204800, 65536).view(256, 256),
              packed.narrow(0, 270336, 256).view(1, 256)]
        (dz1p, dz2p, dz1v, dz2v, dh, _db1p, _db2p, _db3p, _db1v, _db2v,
         _db3v, ws) = ext.mlp_heads_bwd(dlogits, dvalue, stash, wT, A)
        # all six wgrads (dW = dz^T @ act) in one MFMA launch; its block 72
        # also converts the f32 bias-grad partials to contiguous bf16
        (dw1p, dw2p, dw3p, dw1v, dw2v, dw3v, db1p, db2p, db3p, db1v,
         db2v, db3v) = ext.mlp_heads_wgrad(
            dz1p, dz2p, dz1v, dz2v, dlogits, dvalue, stash, ws, A)
        return (dh, dw1p, db1p, dw2p, db2p, dw3p, db3p,
                dw1v, db1v, dw2v, db2v, dw3v, db3v)


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
