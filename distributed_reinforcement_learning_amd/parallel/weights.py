"""Versioned weight publication over shared memory (seqlock).

Replaces the reference's per-actor cross-process variable-assign pulls
(utils.py:6-22 + agent/impala.py:111-112, call site C3 in SURVEY.md §2.4),
which could deliver a torn mix of step-k and step-k+1 weights.

Writer (learner rank 0): bump version to odd, memcpy the flat fp32 parameter
blob, bump to even. Readers (actors) retry until they observe a stable even
version — wait-free for the learner, lock-free for actors, one memcpy per
sync instead of one RPC per variable.
"""

from __future__ import annotations

import struct
import time
from multiprocessing import shared_memory
from typing import Dict, Optional, Tuple

import numpy as np
import torch

_HDR = struct.Struct("<QQQ")  # version, nbytes, global_step
_HDR_SIZE = 64


def _flat_spec(state_dict: Dict[str, torch.Tensor]):
    """Deterministic (name, shape, numel) layout for the flat blob."""
    spec = []
    for k in sorted(state_dict):
        t = state_dict[k]
        spec.append((k, tuple(t.shape), t.numel()))
    return spec


class WeightPublisher:
    def __init__(self, name: str, state_dict: Dict[str, torch.Tensor]):
        self.name = name
        self.spec = _flat_spec(state_dict)
        self.numel = sum(n for _, _, n in self.spec)
        total = _HDR_SIZE + self.numel * 4
        try:
            old = shared_memory.SharedMemory(name=name)
            old.close()
            old.unlink()
        except FileNotFoundError:
            pass
        self.shm = shared_memory.SharedMemory(name=name, create=True,
                                              size=total)
        self.shm.buf[:_HDR_SIZE] = b"\0" * _HDR_SIZE
        self._payload = np.ndarray((self.numel,), dtype=np.float32,
                                   buffer=self.shm.buf, offset=_HDR_SIZE)
        self._version = 0
        self._scratch = torch.empty(self.numel, dtype=torch.float32,
                                    device="cpu", pin_memory=False)

    def publish(self, state_dict: Dict[str, torch.Tensor],
                global_step: int = 0) -> None:
        # gather to one flat CPU tensor (single D2H when params are on GPU)
        off = 0
        for k, shape, n in self.spec:
            t = state_dict[k].detach().reshape(-1).to(
                torch.float32)
            self._scratch[off:off + n].copy_(t, non_blocking=False)
            off += n
        self._version += 1  # odd: write in progress
        struct.pack_into("<QQQ", self.shm.buf, 0, self._version,
                         self.numel * 4, global_step)
        self._payload[:] = self._scratch.numpy()
        self._version += 1  # even: stable
        struct.pack_into("<QQQ", self.shm.buf, 0, self._version,
                         self.numel * 4, global_step)

    def close(self) -> None:
        self._payload = None
        self.shm.close()
        try:
            self.shm.unlink()
        except FileNotFoundError:
            pass


class WeightSubscriber:
    def __init__(self, name: str, state_dict: Dict[str, torch.Tensor],
                 attach_timeout: float = 60.0):
        self.name = name
        self.spec = _flat_spec(state_dict)
        self.numel = sum(n for _, _, n in self.spec)
        deadline = time.time() + attach_timeout
        while True:
            try:
                self.shm = shared_memory.SharedMemory(name=name)
                break
            except FileNotFoundError:
                if time.time() > deadline:
                    raise TimeoutError(f"weights shm {name!r} not published")
                time.sleep(0.05)
        self._payload = np.ndarray((self.numel,), dtype=np.float32,
                                   buffer=self.shm.buf, offset=_HDR_SIZE)
        self.last_version = 0

    def _header(self) -> Tuple[int, int, int]:
        return _HDR.unpack_from(self.shm.buf, 0)

    def wait_for_first(self, timeout: float = 120.0) -> None:
        deadline = time.time() + timeout
        while self._header()[0] == 0:
            if time.time() > deadline:
                raise TimeoutError("no weights published yet")
            time.sleep(0.05)

    def pull(self, state_dict: Dict[str, torch.Tensor],
             max_retries: int = 1000) -> Optional[int]:
        """Copy a consistent snapshot into ``state_dict`` (in place).
        Returns the publisher's global_step, or None if nothing new."""
        v0, _, step = self._header()
        if v0 == self.last_version or v0 == 0:
            return None
        for _ in range(max_retries):
            v0, _, step = self._header()
            if v0 % 2 == 1:
                time.sleep(0.0005)
                continue
            flat = torch.from_numpy(self._payload.copy())
            v1, _, _ = self._header()
            if v0 == v1:
                off = 0
                with torch.no_grad():
                    for k, shape, n in self.spec:
                        state_dict[k].copy_(
                            flat[off:off + n].view(shape))
                        off += n
                self.last_version = v0
                return step
        raise RuntimeError("weight snapshot never stabilized")

    def close(self) -> None:
        self._payload = None
        self.shm.close()
