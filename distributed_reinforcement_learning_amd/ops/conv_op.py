"""Custom MFMA conv stack autograd wrapper (K1, ops/hip/conv.hip).

Layer ids (conv.hip convcfg): 0 = 84x84x4 u8 -> 20x20x32 (IMPALA/A3C/Ape-X),
1 = 84x84x1 u8 (R2D2 POMDP), 2 = 20x20x32 -> 9x9x64, 3 = 9x9x64 -> 7x7x64.

The forward fuses /255-normalize (u8 layers), bias and ReLU; the backward is
relu-mask -> custom wgrad (MFMA, split-M atomics) + dgrad (MFMA implicit
GEMM) + bias column reduction. Weight layout is the channels_last flat view
[CO][KH*KW*CI], built in-graph so grads flow back to the nn.Conv2d weight.
"""

from __future__ import annotations

import torch

from distributed_reinforcement_learning_amd import ops as _ops

_LAYER_CO = {0: 32, 1: 32, 2: 64, 3: 64}

# The weight gradients feed ONLY the optimizer (after the whole backward),
# while dx continues the critical dgrad chain — so the wgrad+finalize pair
# runs on a dedicated side stream and overlaps the rest of the backward
# (~95 us of MFMA wgrad work hidden behind dgrad/heads/embed). Under
# hipGraph capture the fork becomes graph edges and the graph owner joins
# once at capture end (join_wgrad_stream); eagerly each backward joins
# immediately (correctness over overlap — warmup only).
_WGRAD_STREAM = None


def _side_enabled() -> bool:
    # measured SLOWER inside the captured step on ROCm 7.2 (cross-stream
    # graph edges cost more than the overlap wins — 0.59 -> 1.01 ms/step,
    # r2); opt-in for experiments
    import os
    return os.environ.get("DRLA_WGRAD_SIDE") == "1"


def wgrad_stream() -> torch.cuda.Stream:
    global _WGRAD_STREAM
    if _WGRAD_STREAM is None:
        _WGRAD_STREAM = torch.cuda.Stream()
    return _WGRAD_STREAM


def join_wgrad_stream() -> None:
    """Order the current stream after all pending side-stream wgrads —
    call at the END of a captured backward region."""
    if _WGRAD_STREAM is not None:
        torch.cuda.current_stream().wait_stream(_WGRAD_STREAM)


# when True (set by GraphedImpalaStep around its warmup+captures), the u8
# layer-1 forward kernel bundles a pass-through copy of its input and the
# backward consumes THAT — the graphed step's overlapped H2D rewrites the
# static input during the backward otherwise (torn l1 wgrad reads)
STASH_INPUTS = False


class _ConvLayer(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x: torch.Tensor, w_flat: torch.Tensor,
                bias_f32: torch.Tensor, layer: int):
        ext = _ops.require_ext()
        stash = STASH_INPUTS and layer <= 1
        y, x_st = ext.conv_fwd(layer, x.contiguous(), w_flat.contiguous(),
                               bias_f32.contiguous(), stash)
        ctx.save_for_backward(x_st if stash else x, w_flat, y)
        ctx.layer = layer
        return y

    @staticmethod
    def backward(ctx, dy: torch.Tensor):
        x, w_flat, y = ctx.saved_tensors
        ext = _ops.require_ext()
        layer = ctx.layer
        co = _LAYER_CO[layer]
        # mask kernel writes bias-grad partials into a persistent slot
        # buffer; the wgrad finalize that follows sums them -> bf16 dbias
        if not (dy.is_contiguous() or (dy.stride(-1) == 1
                                        and dy.dim() >= 2)):
            dy = dy.contiguous()
        dy_m = ext.relu_mask_bwd(dy, y, co, layer)
        if _side_enabled():
            cur = torch.cuda.current_stream()
            side = wgrad_stream()
            side.wait_stream(cur)
            with torch.cuda.stream(side):
                dw, dbias = ext.conv_wgrad(layer, x, dy_m)
            if not torch.cuda.is_current_stream_capturing():
                # eager: join now, pin allocator lifetimes across streams
                cur.wait_stream(side)
                for t in (dw, dbias):
                    t.record_stream(cur)
                for t in (x, dy_m):
                    t.record_stream(side)
        else:
            dw, dbias = ext.conv_wgrad(layer, x, dy_m)
        dx = None
        if layer >= 2 and ctx.needs_input_grad[0]:
            dx = ext.conv_dgrad(layer, dy_m, w_flat)
        return dx, dw, dbias, None


def conv_layer(x: torch.Tensor, conv: torch.nn.Conv2d,
               layer: int) -> torch.Tensor:
    """x NHWC (u8 for layers 0/1, bf16 for 2/3); returns NHWC bf16 output.

    conv.weight must be bf16 channels_last ([CO][KH][KW][CI] physically) —
    the flat view below is then a no-copy reshape.
    """
    w = conv.weight
    co = w.shape[0]
    w_flat = w.permute(0, 2, 3, 1).reshape(co, -1)
    return _ConvLayer.apply(x, w_flat, conv.bias, layer)


def atari_conv_stack(stack, x_u8_nhwc: torch.Tensor) -> torch.Tensor:
    """Full custom stack: u8 [N,84,84,C] -> flat [N,3136] bf16 features,
    flattened in NHWC order (identical to the torch path in
    models/blocks.AtariConvStack)."""
    ci = x_u8_nhwc.shape[-1]
    l1 = 0 if ci == 4 else 1
    y = conv_layer(x_u8_nhwc, stack.conv1, l1)
    y = conv_layer(y, stack.conv2, 2)
    y = conv_layer(y, stack.conv3, 3)
    return y.reshape(y.shape[0], -1)


def custom_stack_ok(stack, x: torch.Tensor) -> bool:
    return (x.is_cuda and x.dtype == torch.uint8
            and x.shape[1] == 84 and x.shape[2] == 84
            and x.shape[3] in (1, 4)
            and stack.conv1.weight.dtype == torch.bfloat16
            and _ops.available())
