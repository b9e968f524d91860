"""Batch frame ingest: uint8 NHWC -> normalized float (K11 in SURVEY.md §2.5).

The reference normalizes on the host with numpy (agent/impala.py:133) and ships
float32 through feed_dict. Here frames stay uint8 end-to-end (4 bytes -> 1 byte
per pixel on the wire) and the /255 cast runs on-device, fused into one
HBM-friendly pass (and into the conv's load stage on the full custom-conv
path).
"""

from __future__ import annotations

import torch

from distributed_reinforcement_learning_amd import ops as _ops


def normalize_frames(frames_u8: torch.Tensor,
                     out_dtype: torch.dtype = torch.float32) -> torch.Tensor:
    """frames_u8: uint8 tensor (any shape). Returns frames/255 in out_dtype."""
    if frames_u8.is_cuda:
        ext = _ops.require_ext()
        if out_dtype == torch.float32:
            return ext.normalize_frames_f32(frames_u8.contiguous())
        if out_dtype == torch.bfloat16:
            return ext.normalize_frames_bf16(frames_u8.contiguous())
        raise ValueError(f"unsupported out_dtype {out_dtype}")
    return frames_u8.to(out_dtype) / 255.0
