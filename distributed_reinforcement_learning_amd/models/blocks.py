"""Shared model building blocks.

The reference copy-pastes these per model file (attention_CNN / action_embedding
/ fully_connected / lstm in model/actor_critic.py:3-26,
model/impala_actor_critic.py:5-31, model/apex_value.py:4-20,
model/r2d2_lstm.py:4-24); here they are defined once.

Layout convention: public model APIs take states as the reference stores them —
NHWC [N, H, W, C] float (already /255-normalized) — and convert to NCHW
internally for torch convs. The custom HIP conv path (ops/) consumes NHWC
uint8 directly and fuses the /255 normalize into the first conv's load stage.
"""

from __future__ import annotations

from typing import List, Optional, Sequence

import torch
import torch.nn as nn
import torch.nn.functional as F


class AtariConvStack(nn.Module):
    """Conv 8x8/4 -> 32, 4x4/2 -> 64, 3x3/1 -> 64, VALID padding, ReLU,
    flatten.

    84x84xC input -> 20x20x32 -> 9x9x64 -> 7x7x64 -> 3136 features
    (reference model/impala_actor_critic.py:5-10).
    """

    def __init__(self, in_channels: int):
        super().__init__()
        self.conv1 = nn.Conv2d(in_channels, 32, kernel_size=8, stride=4)
        self.conv2 = nn.Conv2d(32, 64, kernel_size=4, stride=2)
        self.conv3 = nn.Conv2d(64, 64, kernel_size=3, stride=1)
        self.out_features = 7 * 7 * 64
        # frames arrive NHWC (that is how the queue stores them); running the
        # whole stack channels_last keeps MIOpen on its native NHWC kernels
        # and removes the batched_transpose launches around every conv
        # (measured: they were ~8% of step GPU time in eager NCHW).
        self.conv1.to(memory_format=torch.channels_last)
        self.conv2.to(memory_format=torch.channels_last)
        self.conv3.to(memory_format=torch.channels_last)

    def forward(self, x_nhwc: torch.Tensor) -> torch.Tensor:
        # GPU hot path: uint8 frames go straight into the hand-written MFMA
        # implicit-GEMM stack (ops/hip/conv.hip) with the /255 normalize,
        # bias and ReLU fused into each layer. The torch path below is the
        # CPU/fallback reference the parity tests compare against.
        if x_nhwc.is_cuda and x_nhwc.dtype == torch.uint8:
            from distributed_reinforcement_learning_amd.ops.conv_op import (
                atari_conv_stack, custom_stack_ok,
            )
            assert custom_stack_ok(self, x_nhwc), (
                "uint8 frames on GPU require the custom conv stack "
                "(bf16 weights + built extension)")
            return atari_conv_stack(self, x_nhwc)
        # NHWC [N,H,W,C] -> logical NCHW with channels_last layout: a view,
        # no copy, no transpose kernel.
        x = x_nhwc.to(self.conv1.weight.dtype).permute(0, 3, 1, 2)
        if x.device.type == "cuda":
            x = x.contiguous(memory_format=torch.channels_last)
        else:
            x = x.contiguous()
        x = F.relu(self.conv1(x))
        x = F.relu(self.conv2(x))
        x = F.relu(self.conv3(x))
        # flatten in NHWC order (TF semantics, reference
        # impala_actor_critic.py:8-9) — identical math on CPU and GPU so
        # weights transfer between learner and actors; on channels_last this
        # permute+flatten is a free view.
        return x.permute(0, 2, 3, 1).flatten(1)


class ActionEmbedding(nn.Module):
    """one-hot(prev_action) -> 256 -> 256, ReLU
    (reference model/impala_actor_critic.py:12-16).

    one-hot @ W + b is a row gather of W — computed as an embedding lookup
    (SURVEY K2): one index_select instead of a one-hot scatter + skinny GEMM.
    ``table`` is stored [num_action, hidden], i.e. the transpose of the
    nn.Linear weight it replaces; identical math.
    """

    def __init__(self, num_action: int, hidden: int = 256):
        super().__init__()
        self.num_action = num_action
        self.table = nn.Parameter(torch.empty(num_action, hidden))
        self.bias1 = nn.Parameter(torch.zeros(hidden))
        nn.init.kaiming_uniform_(self.table, a=5 ** 0.5)
        self.fc2 = nn.Linear(hidden, hidden)
        self.out_features = hidden

    def forward(self, prev_action: torch.Tensor) -> torch.Tensor:
        from distributed_reinforcement_learning_amd.ops import embed_op, available
        t = self.table
        if (t.is_cuda and t.dtype == torch.bfloat16
                and t.shape[1] == 256 and self.fc2.out_features == 256
                and available()):
            return embed_op.fused_action_embed(
                prev_action.long(), t, self.bias1, self.fc2.weight,
                self.fc2.bias)
        x = F.relu(embed_op.embed_lookup(t, prev_action) + self.bias1)
        return F.relu(self.fc2(x))


class MLPHead(nn.Module):
    """Dense stack with ReLU between hidden layers, optional final activation
    (reference model/impala_actor_critic.py:27-31)."""

    def __init__(self, in_features: int, hidden_list: Sequence[int],
                 out_features: int,
                 final_activation: Optional[str] = None):
        super().__init__()
        layers: List[nn.Module] = []
        prev = in_features
        for h in hidden_list:
            layers.append(nn.Linear(prev, h))
            prev = h
        self.hidden = nn.ModuleList(layers)
        self.out = nn.Linear(prev, out_features)
        self.final_activation = final_activation

    def logits(self, x: torch.Tensor) -> torch.Tensor:
        x = x.to(self.out.weight.dtype)
        for layer in self.hidden:
            x = F.relu(layer(x))
        return self.out(x)

    def forward(self, x: torch.Tensor) -> torch.Tensor:
        x = self.logits(x)
        if self.final_activation == "softmax":
            # softmax/log run in fp32 regardless of the compute dtype
            x = F.softmax(x.float(), dim=-1)
        return x


class _AugGateWeight(torch.autograd.Function):
    """[W ; b] as ONE [K+1, 4H] GEMM operand with no copy.

    The fused optimizer (ops/optim.py) lays the cell's weight and bias out
    ADJACENTLY in its flat parameter buffer, so the augmented matrix is an
    as_strided view over the same storage; pairing it with a ones-column on
    the xh input folds the bias add into the GEMM and — the real win — the
    bias GRADIENT into the existing dW GEMM (dW_aug's bias row), deleting
    the separate [N,4H] column-sum reduce from the captured step (~13 us,
    trace r02).

    The augmentation is PAD=8 rows, not 1, so the GEMM's K dimension stays
    8-aligned (odd K forces hipBLASLt off its fast MFMA tilings — measured
    +7 us on the forward). Row K is the bias; rows K+1..K+7 alias whatever
    follows in the flat buffer and meet only the ZERO columns of the xh
    padding, so they contribute nothing and their grad rows are dropped."""

    PAD = 8

    @staticmethod
    def forward(ctx, weight: torch.Tensor, bias: torch.Tensor):
        K, G = weight.shape
        ctx.K = K
        return weight.as_strided((K + _AugGateWeight.PAD, G), (G, 1))

    @staticmethod
    def backward(ctx, d_aug: torch.Tensor):
        return d_aug[:ctx.K], d_aug[ctx.K]


class LSTMCellTF(nn.Module):
    """LSTM cell with TF-LSTMCell semantics (forget_bias=1.0, no peepholes) —
    reference model/impala_actor_critic.py:18-25 runs tf.nn.rnn_cell.LSTMCell
    over a length-1 sequence, i.e. exactly one cell step.

    Gates are computed as one fused GEMM over concat([x, h]) with gate order
    [i, g, f, o]; the elementwise tail (sigmoid/tanh/blend) is the piece the
    HIP fused-gate kernel (ops/hip/lstm_gates.hip) replaces on gfx950.
    """

    def __init__(self, input_size: int, hidden_size: int,
                 forget_bias: float = 1.0):
        super().__init__()
        self.input_size = input_size
        self.hidden_size = hidden_size
        self.forget_bias = forget_bias
        self.weight = nn.Parameter(
            torch.empty(input_size + hidden_size, 4 * hidden_size))
        self.bias = nn.Parameter(torch.zeros(4 * hidden_size))
        nn.init.xavier_uniform_(self.weight)

    def gates_from_xh(self, xh: torch.Tensor) -> torch.Tensor:
        """Pre-concatenated [x, h] input (the batched unroll builds ONE
        cat of conv-features + action-embedding + h instead of two)."""
        if xh.is_cuda and xh.dtype == torch.bfloat16:
            return torch.addmm(self.bias, xh, self.weight)
        return torch.addmm(self.bias.to(xh.dtype), xh, self.weight)

    def _aug_weight_ok(self) -> bool:
        """True when weight||bias sit adjacently in the flat optimizer
        buffer (ops/optim.flatten_dense_params packs them back-to-back
        when the weight's 8-aligned slot has no hole)."""
        w, b = self.weight, self.bias
        if not (w.is_cuda and w.dtype == torch.bfloat16
                and b.dtype == w.dtype and w.numel() % 8 == 0
                and w.data_ptr() + w.numel() * w.element_size()
                == b.data_ptr()):
            return False
        # the augmented view (PAD=8 rows past the weight) must stay inside
        # w's storage (true for the flat optimizer buffer; guards against
        # accidental adjacency of separate allocations)
        need = (w.storage_offset() + w.numel()
                + _AugGateWeight.PAD * w.shape[1]) * w.element_size()
        return w.untyped_storage().nbytes() >= need

    def _ones_col(self, n: int, xh_dtype, dev) -> torch.Tensor:
        """[n, PAD] padding block: column 0 is ones (meets the bias row),
        columns 1..PAD-1 are zeros (annihilate the alias rows)."""
        cache = getattr(self, "_ones_cache", None)
        if cache is None:
            cache = self._ones_cache = {}
        key = (n, xh_dtype, dev)
        if key not in cache:
            pad = torch.zeros(n, _AugGateWeight.PAD, dtype=xh_dtype,
                              device=dev)
            pad[:, 0] = 1
            cache[key] = pad
        return cache[key]

    def forward_cat(self, parts, c: torch.Tensor):
        """Batched-unroll entry: cats the xh parts once (casting h to the
        compute dtype) and runs GEMM + fused tail. With the flat-buffer
        adjacency available, the bias rides the GEMM as an augmented row
        against a ones-column (see _AugGateWeight)."""
        dtype = parts[0].dtype
        parts = [p if p.dtype == dtype else p.to(dtype) for p in parts]
        if self._aug_weight_ok():
            n = parts[0].shape[0]
            xh = torch.cat(parts + [self._ones_col(n, dtype,
                                                   parts[0].device)], dim=1)
            g = torch.mm(xh, _AugGateWeight.apply(self.weight, self.bias))
        else:
            xh = torch.cat(parts, dim=1)
            g = self.gates_from_xh(xh)
        from distributed_reinforcement_learning_amd.ops.lstm_op import (
            lstm_fused_step,
        )
        if g.is_cuda:
            return lstm_fused_step(g, c.float(), self.forget_bias)
        return lstm_fused_step(g.float(), c.float(), self.forget_bias)

    def forward_xh(self, xh: torch.Tensor, c: torch.Tensor):
        from distributed_reinforcement_learning_amd.ops.lstm_op import (
            lstm_fused_step,
        )
        g = self.gates_from_xh(xh)
        if g.is_cuda:
            return lstm_fused_step(g, c.float(), self.forget_bias)
        return lstm_fused_step(g.float(), c.float(), self.forget_bias)

    def gates(self, x: torch.Tensor, h: torch.Tensor) -> torch.Tensor:
        xh = torch.cat([x, h.to(x.dtype)], dim=1)
        # addmm fuses the bias into the GEMM epilogue (one fewer launch);
        # torch's bias-grad reduce (~7 us) beats a custom column sum here —
        # a [640,1024] colsum can fill at most ~16 blocks, measured 29 us
        # (profiles r27), so the reduction stays on torch
        return torch.addmm(self.bias.to(xh.dtype), xh, self.weight)

    def forward(self, x: torch.Tensor, h: torch.Tensor, c: torch.Tensor):
        g = self.gates(x, h)
        # elementwise tail: fused HIP kernel on gfx950 (fp32 states), eager
        # math on CPU — ops/lstm_op.py
        from distributed_reinforcement_learning_amd.ops import lstm_fused_step
        if g.is_cuda:
            # the fused tail consumes bf16 gates natively
            return lstm_fused_step(g, c.float(), self.forget_bias)
        return lstm_fused_step(g.float(), c.float(), self.forget_bias)
