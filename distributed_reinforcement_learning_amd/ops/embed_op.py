"""Action-embedding lookup with custom backward scatter (K2).

Forward is a plain index_select (fast everywhere); on GPU the backward uses
the drla scatter kernel instead of torch's embedding_backward_feature_kernel
(38us -> ~5us per step at the reference shape)."""

from __future__ import annotations

import torch

from distributed_reinforcement_learning_amd import ops as _ops


class _EmbedLookup(torch.autograd.Function):
    @staticmethod
    def forward(ctx, table: torch.Tensor, idx: torch.Tensor):
        ctx.save_for_backward(idx)
        ctx.num_rows = table.shape[0]
        ctx.want_bf16 = table.dtype == torch.bfloat16
        return table.index_select(0, idx)

    @staticmethod
    def backward(ctx, grad_out: torch.Tensor):
        (idx,) = ctx.saved_tensors
        ext = _ops.require_ext()
        grad_table = ext.embed_bwd(idx.contiguous(),
                                   grad_out.contiguous(), ctx.num_rows,
                                   ctx.want_bf16)
        return grad_table, None


def embed_lookup(table: torch.Tensor, idx: torch.Tensor) -> torch.Tensor:
    """table [A,H], idx [N] long -> [N,H]; custom backward on GPU."""
    if table.is_cuda:
        return _EmbedLookup.apply(table, idx.long())
    return torch.nn.functional.embedding(idx.long(), table)
