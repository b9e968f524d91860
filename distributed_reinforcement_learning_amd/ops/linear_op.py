"""Linear/bias autograd helpers with custom bf16 bias-grad reduction.

torch's bf16 column reduce_kernel costs ~9 us per bias grad at the
flagship shapes ([640,1024] gates, [640,256] embedding layers); the
drla_colsum_bf16 kernel is load-bound (~3 us). Used by LSTMCellTF.gates
and ActionEmbedding on the GPU path.
"""

from __future__ import annotations

import torch

from distributed_reinforcement_learning_amd import ops as _ops


class _AddmmColsumBias(torch.autograd.Function):
    """y = bias + x @ W  (W [in,out] as LSTMCellTF stores it)."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, bias: torch.Tensor):
        ctx.save_for_backward(x, w)
        return torch.addmm(bias, x, w)

    @staticmethod
    def backward(ctx, g: torch.Tensor):
        x, w = ctx.saved_tensors
        ext = _ops.require_ext()
        g = g.contiguous()
        dx = g.mm(w.t())
        dw = x.t().mm(g)
        db = ext.colsum_bf16(g)
        return dx, dw, db


class _LinearColsumBias(torch.autograd.Function):
    """y = x @ W^T + bias  (W [out,in], the nn.Linear convention)."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, w: torch.Tensor, bias: torch.Tensor):
        ctx.save_for_backward(x, w)
        return torch.addmm(bias, x, w.t())

    @staticmethod
    def backward(ctx, g: torch.Tensor):
        x, w = ctx.saved_tensors
        ext = _ops.require_ext()
        g = g.contiguous()
        dx = g.mm(w)
        dw = g.t().mm(x)
        db = ext.colsum_bf16(g)
        return dx, dw, db


class _AddBiasColsum(torch.autograd.Function):
    """y = x + bias (row broadcast); backward passes dx through untouched
    and reduces db with the custom colsum."""

    @staticmethod
    def forward(ctx, x: torch.Tensor, bias: torch.Tensor):
        return x + bias

    @staticmethod
    def backward(ctx, g: torch.Tensor):
        ext = _ops.require_ext()
        return g, ext.colsum_bf16(g.contiguous())


def addmm_colsum(x, w, bias):
    return _AddmmColsumBias.apply(x, w, bias)


def linear_colsum(x, w, bias):
    return _LinearColsumBias.apply(x, w, bias)


def add_bias_colsum(x, bias):
    return _AddBiasColsum.apply(x, bias)
