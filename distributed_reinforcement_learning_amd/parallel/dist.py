"""Learner data-parallelism over RCCL/xGMI.

One process per GPU; ``torch.distributed`` backend "nccl" (RCCL on ROCm) for
GPU runs, "gloo" for CPU tests. Gradients live in ONE flat contiguous buffer
(ops/optim.py), so the per-step all-reduce is a single fused RCCL call — at
reference model sizes (~16 MB of grads) the all-reduce is latency-bound and
one bucket beats any bucketing schedule (SURVEY §2.4 xGMI note).
"""

from __future__ import annotations

import datetime
import os
from typing import Optional

import torch
import torch.distributed as dist


def is_distributed() -> bool:
    return dist.is_available() and dist.is_initialized()


def world_size() -> int:
    return dist.get_world_size() if is_distributed() else 1


def rank() -> int:
    return dist.get_rank() if is_distributed() else 0


def init_distributed(backend: Optional[str] = None,
                     timeout_s: float = 300.0) -> int:
    """Initialize from torchrun-style env vars; no-op at WORLD_SIZE<=1.
    Returns the local rank."""
    ws = int(os.environ.get("WORLD_SIZE", "1"))
    local_rank = int(os.environ.get("LOCAL_RANK", os.environ.get("RANK", "0")))
    if ws <= 1:
        return local_rank
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if not dist.is_initialized():
        dist.init_process_group(
            backend=backend,
            timeout=datetime.timedelta(seconds=timeout_s))
    if torch.cuda.is_available():
        torch.cuda.set_device(local_rank)
    return local_rank


class FlatAllReducer:
    """Average one flat gradient buffer across ranks with a single
    all-reduce call."""

    def __init__(self, flat_grads: torch.Tensor):
        self.flat_grads = flat_grads
        self._inv_world = 1.0 / world_size()
        # RCCL fuses the divide into the collective (ReduceOp.AVG);
        # gloo (CPU tests) lacks AVG -> sum + scale fallback
        self._use_avg = (is_distributed()
                         and dist.get_backend() == "nccl")

    def all_reduce(self) -> None:
        if not is_distributed():
            return
        if self._use_avg:
            dist.all_reduce(self.flat_grads, op=dist.ReduceOp.AVG)
        else:
            dist.all_reduce(self.flat_grads, op=dist.ReduceOp.SUM)
            self.flat_grads.mul_(self._inv_world)


def handle_capture_failure(exc: BaseException) -> None:
    """Policy for a failed hipGraph capture of the distributed optimizer
    step (gather + RCCL all-reduce + update). A rank that silently falls
    back to an eager all-reduce while its peers replay a captured one
    deadlocks the lockstep replay, so the default is
    FAIL FAST with the traceback. Set ``DRLA_ALLOW_EAGER_REDUCE=1`` to
    accept the eager path (~0.4 ms/step of host latency, measured) — only
    sound when EVERY rank takes it, e.g. a single-process debug run."""
    import sys
    import traceback
    traceback.print_exc()
    if os.environ.get("DRLA_ALLOW_EAGER_REDUCE") == "1":
        print("[drla] WARNING: distributed optimizer-graph capture failed "
              "(traceback above); DRLA_ALLOW_EAGER_REDUCE=1 -> continuing "
              "with the eager all-reduce. All ranks must take this path.",
              file=sys.stderr, flush=True)
        return
    print("[drla] FATAL: hipGraph capture of the distributed optimizer "
          "step failed (traceback above). Refusing the silent eager "
          "fallback; set DRLA_ALLOW_EAGER_REDUCE=1 to accept it on every "
          "rank.", file=sys.stderr, flush=True)
    raise exc


def broadcast_module(module: torch.nn.Module, src: int = 0) -> None:
    """Rank-0 init + broadcast (fixes the reference's re-init race, C5 in
    SURVEY.md §2.4). Strided params (channels_last conv weights re-homed
    into the flat optimizer buffer) bounce through a contiguous staging
    tensor — collectives require dense tensors."""
    if not is_distributed():
        return
    with torch.no_grad():
        for t in module.state_dict().values():
            if t.is_contiguous():
                dist.broadcast(t, src=src)
            else:
                tmp = t.contiguous()
                dist.broadcast(tmp, src=src)
                t.copy_(tmp)
