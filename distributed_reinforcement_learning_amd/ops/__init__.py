"""HIP/CDNA4 kernel bindings (gfx950).

The extension ``_drla_hip`` is built in-tree (setup.py build_ext --inplace,
PYTORCH_ROCM_ARCH=gfx950) from the sources in ops/hip/. Policy:

* On CPU tensors every op falls back to its PyTorch reference implementation
  (the same code the parity tests compare against).
* On CUDA(=HIP) tensors the extension is REQUIRED: if it failed to import, the
  op raises instead of silently running an eager fallback — a GPU run must
  exercise the native kernels.
"""

from __future__ import annotations

import os

_EXT = None
_IMPORT_ERROR: Exception | None = None

try:
    from distributed_reinforcement_learning_amd.ops import _drla_hip as _EXT  # type: ignore
except ImportError as e:  # extension not built (CPU-only dev) — fallbacks run
    _IMPORT_ERROR = e


def available() -> bool:
    return _EXT is not None


def require_ext() -> "object":
    if _EXT is None:
        raise RuntimeError(
            "distributed_reinforcement_learning_amd._drla_hip is not built "
            "but a GPU tensor reached a custom op. Build it in-tree with "
            "`python setup.py build_ext --inplace` (PYTORCH_ROCM_ARCH=gfx950)."
            f" Original import error: {_IMPORT_ERROR}")
    return _EXT


from distributed_reinforcement_learning_amd.ops.vtrace_op import vtrace_scan
from distributed_reinforcement_learning_amd.ops.vtrace_loss_op import fused_vtrace_loss
from distributed_reinforcement_learning_amd.ops.preprocess import normalize_frames
from distributed_reinforcement_learning_amd.ops.lstm_op import lstm_fused_step
from distributed_reinforcement_learning_amd.ops.optim import FusedRMSProp, FusedAdam

__all__ = [
    "available", "require_ext", "vtrace_scan", "fused_vtrace_loss",
    "normalize_frames", "lstm_fused_step", "FusedRMSProp", "FusedAdam",
]
