"""Lock-free shared-memory trajectory transport (replaces reference
distributed_queue/buffer_queue.py's TF FIFOQueues + gRPC, call sites C1/C2 in
SURVEY.md §2.4).

Design: one SPSC ring per actor in POSIX shared memory. The actor (single
producer) writes a fixed-layout trajectory slot and publishes it by bumping
``tail``; the learner (single consumer) drains by bumping ``head``. No locks,
no serialization — fields are numpy views straight into the shm buffer, and
frames travel as uint8 (the reference's A3C/Ape-X queues shipped int32 pixels,
4 bytes/pixel — buffer_queue.py:16; fixed here as SURVEY §7 directs).

``TrajectoryQueue`` aggregates this learner's shard of rings behind the
reference queue API: ``append_to_queue(task, **fields)`` /
``sample_batch(batch)`` / ``get_size()``. With multiple learner ranks, actor
``i`` feeds rank ``i % world_size`` (trajectory scatter).

Each ring header also carries the producer's heartbeat timestamp
(failure detection — SURVEY §5.3).

Memory-ordering contract (x86-64 ONLY): the producer publishes a slot with a
plain u64 ``tail`` store AFTER the payload stores, and the consumer reads
``tail`` before the payload — correct under x86 TSO (stores retire in program
order, loads are not reordered with older loads), which is the only host ISA
this framework targets (MI355X nodes are EPYC hosts). On a weakly-ordered
host (aarch64) the payload/tail ordering would need explicit release/acquire
fences; none are emitted. tests/tsan/ring_tsan.cc carries a C++ re-statement
of this protocol checked under ThreadSanitizer.
"""

from __future__ import annotations

import struct
import time
from multiprocessing import shared_memory
from typing import Dict, Optional, Tuple

import numpy as np

Schema = Dict[str, Tuple[Tuple[int, ...], np.dtype]]

_HDR = struct.Struct("<QQdQQ")  # tail, head, heartbeat, capacity, slot_size
_HDR_SIZE = 64  # padded to a cacheline


def _slot_layout(schema: Schema):
    offsets = {}
    off = 0
    for name, (shape, dtype) in schema.items():
        dtype = np.dtype(dtype)
        nbytes = int(np.prod(shape)) * dtype.itemsize
        # 8-byte align each field
        off = (off + 7) & ~7
        offsets[name] = (off, shape, dtype)
        off += nbytes
    return offsets, (off + 7) & ~7


class TrajectoryRing:
    """Single-producer single-consumer ring over one shm segment."""

    def __init__(self, name: str, schema: Schema, capacity: int,
                 create: bool, attach_timeout: float = 60.0):
        self.name = name
        self.schema = dict(schema)
        self.capacity = int(capacity)
        self.offsets, self.slot_size = _slot_layout(self.schema)
        total = _HDR_SIZE + self.capacity * self.slot_size
        if create:
            try:
                # clean up a stale segment from a dead previous run
                old = shared_memory.SharedMemory(name=name)
                old.close()
                old.unlink()
            except FileNotFoundError:
                pass
            self.shm = shared_memory.SharedMemory(name=name, create=True,
                                                  size=total)
            self.shm.buf[:_HDR_SIZE] = b"\0" * _HDR_SIZE
            _HDR.pack_into(self.shm.buf, 0, 0, 0, time.time(),
                           self.capacity, self.slot_size)
        else:
            deadline = time.time() + attach_timeout
            while True:
                try:
                    self.shm = shared_memory.SharedMemory(name=name)
                    break
                except FileNotFoundError:
                    if time.time() > deadline:
                        raise TimeoutError(
                            f"ring {name!r} not created within "
                            f"{attach_timeout}s")
                    time.sleep(0.05)
        self._created = create
        self._buf = self.shm.buf
        self._ctr = np.ndarray((2,), dtype=np.uint64, buffer=self._buf,
                               offset=0)  # [tail, head]

    # -- header accessors ----------------------------------------------------

    @property
    def tail(self) -> int:
        return int(self._ctr[0])

    @property
    def head(self) -> int:
        return int(self._ctr[1])

    def size(self) -> int:
        return self.tail - self.head

    def heartbeat(self) -> float:
        return _HDR.unpack_from(self._buf, 0)[2]

    def _touch(self) -> None:
        struct.pack_into("<d", self._buf, 16, time.time())

    # -- producer ------------------------------------------------------------

    def _slot_views(self, slot_idx: int) -> Dict[str, np.ndarray]:
        base = _HDR_SIZE + slot_idx * self.slot_size
        return {
            name: np.ndarray(shape, dtype=dtype, buffer=self._buf,
                             offset=base + off)
            for name, (off, shape, dtype) in self.offsets.items()
        }

    def try_push(self, fields: Dict[str, np.ndarray]) -> bool:
        if self.size() >= self.capacity:
            return False
        views = self._slot_views(self.tail % self.capacity)
        for name, view in views.items():
            arr = np.asarray(fields[name], dtype=view.dtype)
            view[...] = arr.reshape(view.shape)
        self._touch()
        # publish: single u64 store after the payload writes (x86 TSO)
        self._ctr[0] = np.uint64(self.tail + 1)
        return True

    def push(self, fields: Dict[str, np.ndarray],
             block: bool = True, poll: float = 0.001) -> bool:
        while not self.try_push(fields):
            if not block:
                return False
            time.sleep(poll)
        return True

    # -- consumer ------------------------------------------------------------

    def try_pop_into(self, out: Dict[str, np.ndarray], row: int) -> bool:
        """Pop one slot into row ``row`` of ``out``. Fields absent from
        ``out`` are skipped (e.g. the IMPALA learner's pinned staging has no
        next_state buffer — the batched unroll never reads it)."""
        if self.size() == 0:
            return False
        views = self._slot_views(self.head % self.capacity)
        for name, view in views.items():
            dst = out.get(name)
            if dst is not None:
                dst[row] = view
        self._ctr[1] = np.uint64(self.head + 1)
        return True

    def close(self, unlink: Optional[bool] = None) -> None:
        self._ctr = None
        self._buf = None
        self.shm.close()
        if unlink if unlink is not None else self._created:
            try:
                self.shm.unlink()
            except FileNotFoundError:
                pass


# ---------------------------------------------------------------------------


def queue_schema_for(algorithm: str, cfg) -> Schema:
    """Per-algorithm trajectory layouts (reference buffer_queue.py §2.2),
    with uint8 frames everywhere."""
    H, W, C = cfg.model_input if len(cfg.model_input) == 3 else (0, 0, 0)
    A = cfg.model_output
    if algorithm == "a3c":
        T = cfg.trajectory
        shape = tuple(cfg.model_input)
        sdt = np.uint8 if len(cfg.model_input) == 3 else np.float32
        return {
            "state": ((T, *shape), sdt),
            "next_state": ((T, *shape), sdt),
            "previous_action": ((T,), np.int32),
            "action": ((T,), np.int32),
            "reward": ((T,), np.float32),
            "done": ((T,), np.bool_),
        }
    if algorithm == "impala":
        T = cfg.trajectory
        L = cfg.lstm_size
        return {
            "state": ((T, H, W, C), np.uint8),
            "next_state": ((T, H, W, C), np.uint8),
            "previous_action": ((T,), np.int32),
            "action": ((T,), np.int32),
            "reward": ((T,), np.float32),
            "done": ((T,), np.bool_),
            "behavior_policy": ((T, A), np.float32),
            "initial_h": ((T, L), np.float32),
            "initial_c": ((T, L), np.float32),
        }
    if algorithm == "apex":
        T = cfg.trajectory
        return {
            "state": ((T, H, W, C), np.uint8),
            "next_state": ((T, H, W, C), np.uint8),
            "previous_action": ((T,), np.int32),
            "action": ((T,), np.int32),
            "reward": ((T,), np.float32),
            "done": ((T,), np.bool_),
        }
    if algorithm == "r2d2":
        L = cfg.seq_len
        Hs = cfg.lstm_size
        return {
            "state": ((L, H, W, C), np.uint8),
            "previous_action": ((L,), np.int32),
            "action": ((L,), np.int32),
            "reward": ((L,), np.float32),
            "done": ((L,), np.bool_),
            "initial_h": ((L, Hs), np.float32),
            "initial_c": ((L, Hs), np.float32),
        }
    raise KeyError(algorithm)


class TrajectoryQueue:
    """This learner rank's view over its shard of actor rings."""

    def __init__(self, schema: Schema, num_actors: int, queue_size: int,
                 *, role: str, namespace: str, actor_task: int = -1,
                 rank: int = 0, world_size: int = 1,
                 ring_capacity: Optional[int] = None):
        self.schema = schema
        self.num_actors = num_actors
        self.namespace = namespace
        cap = ring_capacity or max(2, queue_size // max(1, num_actors))
        self.role = role
        if role == "learner":
            self.actor_ids = [i for i in range(num_actors)
                              if i % world_size == rank]
            self.rings = {
                i: TrajectoryRing(self._ring_name(i), schema, cap,
                                  create=True)
                for i in self.actor_ids
            }
            self._rr = 0
        elif role == "actor":
            assert 0 <= actor_task < num_actors
            self.actor_ids = [actor_task]
            self.rings = {
                actor_task: TrajectoryRing(self._ring_name(actor_task),
                                           schema, cap, create=False)
            }
        else:
            raise ValueError(role)

    def _ring_name(self, actor_id: int) -> str:
        return f"drla_{self.namespace}_r{actor_id}"

    # -- actor side ----------------------------------------------------------

    def append_to_queue(self, task: int, block: bool = True,
                        **fields) -> bool:
        return self.rings[task].push(fields, block=block)

    # -- learner side --------------------------------------------------------

    def get_size(self) -> int:
        return sum(r.size() for r in self.rings.values())

    def sample_batch(self, batch_size: int, timeout: Optional[float] = None,
                     poll: float = 0.001,
                     out: Optional[Dict[str, np.ndarray]] = None
                     ) -> Dict[str, np.ndarray]:
        """Blocking round-robin gather of ``batch_size`` trajectories into
        stacked arrays (batch dim first). Pass ``out`` (e.g. numpy views of
        the graphed step's pinned staging buffers) to fill caller-owned
        memory with zero extra copies."""
        if out is None:
            out = {
                name: np.empty((batch_size, *shape), dtype=dtype)
                for name, (shape, dtype) in self.schema.items()
            }
        ids = list(self.rings)
        filled = 0
        deadline = time.time() + timeout if timeout else None
        while filled < batch_size:
            progressed = False
            for k in range(len(ids)):
                ring = self.rings[ids[(self._rr + k) % len(ids)]]
                if ring.try_pop_into(out, filled):
                    filled += 1
                    progressed = True
                    if filled == batch_size:
                        break
            self._rr = (self._rr + 1) % len(ids)
            if not progressed:
                if deadline and time.time() > deadline:
                    raise TimeoutError(
                        f"sample_batch: only {filled}/{batch_size} after "
                        f"{timeout}s")
                time.sleep(poll)
        return out

    def heartbeats(self) -> Dict[int, float]:
        return {i: r.heartbeat() for i, r in self.rings.items()}

    def close(self) -> None:
        for r in self.rings.values():
            r.close()
