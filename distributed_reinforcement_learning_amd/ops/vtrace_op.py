"""Fused V-trace reverse scan (K5 in SURVEY.md §2.5).

vs_minus_v[t] = delta[t] + discount[t] * c[t] * vs_minus_v[t+1], scanned in
reverse over T with all B lanes parallel — one kernel launch instead of the
reference's T-step tf.scan graph (vtrace.py:88-100).
"""

from __future__ import annotations

import torch

from distributed_reinforcement_learning_amd import ops as _ops


def vtrace_scan(deltas: torch.Tensor, discounts: torch.Tensor,
                cs: torch.Tensor) -> torch.Tensor:
    """All inputs [B, T] float32. Returns vs_minus_v [B, T]."""
    if deltas.is_cuda:
        ext = _ops.require_ext()
        return ext.vtrace_scan(deltas.contiguous(), discounts.contiguous(),
                               cs.contiguous())
    B, T = deltas.shape
    acc = torch.zeros(B, dtype=deltas.dtype, device=deltas.device)
    out = torch.empty_like(deltas)
    for t in reversed(range(T)):
        acc = deltas[:, t] + discounts[:, t] * cs[:, t] * acc
        out[:, t] = acc
    return out
