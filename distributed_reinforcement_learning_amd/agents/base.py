"""Shared agent machinery: device/dtype policy, polynomial LR decay,
reward clipping, checkpointing, weight sync wiring.

The reference's Agent classes (agent/{a3c,impala,apex,r2d2}.py) each rebuild
this; here it is one base. Checkpoints keep the reference's method names
(save_weights/load_weights — agent/impala.py:105-109) with a torch.save dict
{model, optimizer, global_step} as the layout.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np
import torch

from distributed_reinforcement_learning_amd.parallel.dist import (
    FlatAllReducer, is_distributed,
)


def polynomial_decay(start: float, end: float, step: int,
                     horizon: int) -> float:
    """tf.train.polynomial_decay(power=1) semantics
    (reference agent/impala.py:96)."""
    frac = min(step, horizon) / float(horizon)
    return (start - end) * (1.0 - frac) + end


def clip_rewards(r: torch.Tensor, mode: str) -> torch.Tensor:
    """abs_one / soft_asymmetric / none (reference agent/impala.py:45-49)."""
    if mode == "abs_one":
        return torch.clamp(r, -1.0, 1.0)
    if mode == "soft_asymmetric":
        squeezed = torch.tanh(r / 5.0)
        return torch.where(r < 0, 0.3 * squeezed, squeezed) * 5.0
    if mode == "none":
        return r
    raise ValueError(mode)


class AgentBase:
    """Device/dtype policy + LR schedule + checkpoint + sync plumbing."""

    def __init__(self, device: str = "cpu",
                 compute_dtype: torch.dtype = torch.bfloat16):
        self.device = torch.device(device)
        self.compute_dtype = compute_dtype
        self.global_step = 0
        self.num_env_frames = 0
        # wired by trainers (parallel/weights.py): learner gets a publisher,
        # actors get a subscriber
        self.weight_publisher = None
        self.weight_subscriber = None
        self._all_reducer: Optional[FlatAllReducer] = None

    # -- dtype policy ----------------------------------------------------------
    # GPU learners run bf16-NATIVE models (weights stored bf16, fp32 master in
    # the fused optimizer) — no autocast, no per-layer weight casts. CPU
    # actors/learners stay fp32.

    @property
    def model_dtype(self) -> torch.dtype:
        if self.device.type == "cuda" and self.compute_dtype == torch.bfloat16:
            return torch.bfloat16
        return torch.float32

    def finalize_model(self, model: torch.nn.Module) -> torch.nn.Module:
        model = model.to(self.device)
        if self.model_dtype != torch.float32:
            model = model.to(self.model_dtype)
        return model

    def autocast(self):
        # kept for API compatibility: bf16 is native now, nothing to autocast
        import contextlib
        return contextlib.nullcontext()

    # -- data movement -------------------------------------------------------

    def to_device(self, arr, dtype=None) -> torch.Tensor:
        t = arr if isinstance(arr, torch.Tensor) \
            else torch.as_tensor(np.asarray(arr))
        if dtype is not None:
            t = t.to(dtype)
        return t.to(self.device, non_blocking=True)

    def frames_to_device(self, frames) -> torch.Tensor:
        """uint8 (or float) frames -> normalized float32 on device.

        uint8 stays uint8 across the H2D copy (1 byte/pixel on the bus); the
        /255 cast runs on-device (ops/preprocess.py, HIP kernel on gfx950) or
        fused into conv layer 1 (custom MFMA stack). Device tensors (e.g.
        from the GPU replay shard) pass straight through. Float inputs
        (vector obs) pass through unscaled.
        """
        t = frames if isinstance(frames, torch.Tensor) \
            else torch.as_tensor(np.asarray(frames))
        if t.dtype == torch.uint8:
            t = t.to(self.device, non_blocking=True)
            return self.prepare_frames(t)
        return t.to(self.device, dtype=torch.float32, non_blocking=True)

    def prepare_frames(self, t_u8: torch.Tensor) -> torch.Tensor:
        """Device-side frame prep. With the custom MFMA conv stack active
        (bf16 GPU model, 84x84 u8 frames) the /255 normalize is fused into
        conv layer 1, so frames stay uint8; otherwise normalize here."""
        if (t_u8.is_cuda and self.model_dtype == torch.bfloat16
                and t_u8.shape[-3:-1] == (84, 84)
                and t_u8.shape[-1] in (1, 4)):
            from distributed_reinforcement_learning_amd import ops as _o
            if _o.available():
                return t_u8
        from distributed_reinforcement_learning_amd.ops import normalize_frames
        return normalize_frames(t_u8, out_dtype=self.model_dtype)

    # -- model plumbing (subclasses set self.model / self.optimizer) ---------

    model: torch.nn.Module
    optimizer = None

    def lr_at(self, step: int) -> float:
        return polynomial_decay(self.start_learning_rate,
                                self.end_learning_rate, step,
                                self.learning_frame)

    def setup_all_reduce(self) -> None:
        if is_distributed():
            # the rank-0 broadcast that precedes this rewrote the bf16
            # model copy out-of-band; the fp32 master must follow or the
            # first update reverts the broadcast and diverges the ranks
            if self.optimizer is not None:
                self.optimizer.refresh_master()
            self._all_reducer = FlatAllReducer(self.optimizer.flat_grads)

    def reduce_gradients(self) -> None:
        if self._all_reducer is not None:
            self._all_reducer.all_reduce()

    # -- weight sync ---------------------------------------------------------

    def parameter_sync(self) -> Optional[int]:
        """Actor side: pull the newest published weights (replaces reference
        copy_src_to_dst assigns, utils.py:6-22). Returns publisher step or
        None."""
        if self.weight_subscriber is None:
            return None
        # state_dict() returns references to the live tensors; pull copies
        # the snapshot into them in place — no load_state_dict round trip
        sd = self.model.state_dict()
        return self.weight_subscriber.pull(sd)

    def publish_weights(self) -> None:
        if self.weight_publisher is not None:
            self.weight_publisher.publish(self.model.state_dict(),
                                          global_step=self.global_step)

    # -- checkpointing -------------------------------------------------------

    def _checkpoint_extra(self) -> Dict:
        return {}

    def save_weights(self, path: str) -> None:
        blob = {
            "model": self.model.state_dict(),
            "optimizer": (self.optimizer.state_dict()
                          if self.optimizer is not None else None),
            "global_step": self.global_step,
            "num_env_frames": self.num_env_frames,
        }
        blob.update(self._checkpoint_extra())
        torch.save(blob, path)

    def load_weights(self, path: str) -> None:
        blob = torch.load(path, map_location=self.device,
                          weights_only=False)
        self.model.load_state_dict(blob["model"])
        if self.optimizer is not None and blob.get("optimizer") is not None:
            self.optimizer.load_state_dict(blob["optimizer"])
        elif self.optimizer is not None:
            # checkpoint without optimizer state: re-derive the fp32
            # master from the restored model copy
            self.optimizer.refresh_master()
        self.global_step = blob.get("global_step", 0)
        self.num_env_frames = blob.get("num_env_frames", 0)
        self._load_checkpoint_extra(blob)

    def _load_checkpoint_extra(self, blob: Dict) -> None:
        pass
