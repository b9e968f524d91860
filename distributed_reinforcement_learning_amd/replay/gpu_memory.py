"""GPU-resident prioritized replay (K10) — replay shards sized for 288 GB
HBM per GPU (BASELINE.json).

Priorities, the segment tree AND the sample payloads live on-device:
TD errors from the loss kernels feed ``update_batch`` without ever visiting
the host, ``sample`` draws its stratified offsets with device RNG (no sync),
and ``gather`` returns index_select views the learner trains on directly.
Semantics mirror replay/memory.py (reference buffer_queue.py:373-416):
priority (|err|+e)^a, stratified segments, IS weights normalized by max,
beta 0.4 -> 1.0 by +0.001 per sample call.

Per-GPU sharding: with one learner rank per GPU, each rank owns its own
GpuMemory fed by its own actor shard (parallel/queue.py) — capacity 1e5
Ape-X transitions is ~5.7 GB of uint8 frames, <2% of one MI355X's HBM.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import torch

from distributed_reinforcement_learning_amd import ops as _ops


class GpuMemory:
    e = 0.001
    a = 0.6
    beta_start = 0.4
    beta_increment_per_sampling = 0.001

    def __init__(self, capacity: int, fields: Dict[str, Tuple[tuple,
                                                              torch.dtype]],
                 device: str = "cuda:0", seed: Optional[int] = None):
        self.capacity = int(capacity)
        self.device = torch.device(device)
        self.tree = torch.zeros(2 * self.capacity - 1, dtype=torch.float32,
                                device=self.device)
        self.data = {
            name: torch.zeros((self.capacity, *shape), dtype=dtype,
                              device=self.device)
            for name, (shape, dtype) in fields.items()
        }
        self.write = 0
        self.n_entries = 0
        self.beta = self.beta_start
        # β and n_entries live in device buffers so the sample math is
        # hipGraph-capturable (the host advances them; replays read them)
        self.beta_buf = torch.tensor([self.beta], device=device)
        self.n_entries_buf = torch.zeros(1, device=device)
        # sampling uses the DEFAULT cuda generator: custom Generator
        # objects are not registered with hipGraph capture ("Attempt to
        # increase offset for a CUDA generator not in capture mode"); the
        # default generator's philox offsets are graph-managed
        self.gen = None
        if seed is not None:
            torch.cuda.manual_seed(seed)

    def _prio(self, errors: torch.Tensor) -> torch.Tensor:
        return (errors.abs().float() + self.e) ** self.a

    @torch.no_grad()
    def add_batch(self, errors: torch.Tensor,
                  samples: Dict[str, torch.Tensor]) -> None:
        """errors [n] (device); samples name -> [n, ...] device tensors."""
        ext = _ops.require_ext()
        n = errors.numel()
        rows = (self.write + torch.arange(n, device=self.device)) \
            % self.capacity
        for k, buf in self.data.items():
            buf[rows] = samples[k].to(buf.dtype)
        idxs = rows + (self.capacity - 1)
        ext.per_update(self.tree, idxs.contiguous(),
                       self._prio(errors).contiguous(), self.capacity)
        self.write = int((self.write + n) % self.capacity)
        self.n_entries = min(self.n_entries + n, self.capacity)
        self.n_entries_buf.fill_(float(self.n_entries))

    @torch.no_grad()
    def sample(self, n: int):
        """Returns (rows [n] i64, tree idxs [n] i64, is_weight [n] f32),
        all on-device; no host sync."""
        ext = _ops.require_ext()
        self.advance_beta()
        return self.sample_static(n)

    def advance_beta(self) -> None:
        """Host-side β annealing (reference buffer_queue.py:398): call once
        per sample — the graphed learner calls it before each replay."""
        self.beta = min(1.0, self.beta + self.beta_increment_per_sampling)
        self.beta_buf.fill_(self.beta)

    @torch.no_grad()
    def sample_static(self, n: int):
        """Capture-safe sample body: every input that changes over training
        (β, n_entries, priorities) is read from device memory."""
        ext = _ops.require_ext()
        total = self.tree[0]
        u = torch.rand(n, device=self.device)
        s = (torch.arange(n, device=self.device, dtype=torch.float32) + u) \
            * (total / n)
        idxs, prios = ext.per_sample(self.tree, s.contiguous(),
                                     self.n_entries_buf, self.capacity)
        probs = prios / total
        w = (self.n_entries_buf * probs).pow(-self.beta_buf)
        w = w / w.max()
        rows = idxs - (self.capacity - 1)
        return rows, idxs, w

    def _ensure_gather(self, n: int) -> None:
        """Persistent gather outputs + device descriptor tables for
        drla_multi_gather (one kernel replaces the 7-launch index_select
        chain). Built lazily per batch size, BEFORE any hipGraph capture
        (the graphed learner's eager warmup triggers it)."""
        if getattr(self, "_g_n", None) == n:
            return
        out = {k: torch.empty((n, *buf.shape[1:]), dtype=buf.dtype,
                              device=self.device)
               for k, buf in self.data.items()}
        fbytes = [buf.element_size() * int(buf[0].numel()) if buf.dim() > 1
                  else buf.element_size()
                  for buf in self.data.values()]
        i64 = dict(dtype=torch.int64, device=self.device)
        self._g_out = out
        self._g_srcs = torch.tensor(
            [buf.data_ptr() for buf in self.data.values()], **i64)
        self._g_dsts = torch.tensor(
            [t.data_ptr() for t in out.values()], **i64)
        self._g_fbytes = torch.tensor(fbytes, **i64)
        self._g_maxchunks = max((fb + 15) // 16 for fb in fbytes)
        self._g_n = n

    @torch.no_grad()
    def gather(self, rows: torch.Tensor) -> Dict[str, torch.Tensor]:
        """All fields of the sampled rows in ONE kernel; returns views of
        persistent per-batch-size output buffers (callers consume them
        within the step — the graphed learner overwrites them per replay,
        exactly like its other static buffers)."""
        ext = _ops.require_ext()
        n = rows.numel()
        self._ensure_gather(n)
        ext.multi_gather(rows.contiguous(), self._g_srcs, self._g_dsts,
                         self._g_fbytes, self._g_maxchunks)
        return dict(self._g_out)

    @torch.no_grad()
    def update_batch(self, idxs: torch.Tensor,
                     errors: torch.Tensor) -> None:
        ext = _ops.require_ext()
        ext.per_update(self.tree, idxs.contiguous(),
                       self._prio(errors).contiguous(), self.capacity)

    @torch.no_grad()
    def rebuild(self) -> None:
        """Recompute every interior sum from the leaves (~log2(cap)
        launches). The float32 atomicAdd delta chains in drla_per_update
        drift the interior sums over millions of updates (the CPU twin
        replay/sum_tree.py carries float64); the graphed learner calls
        this every few thousand steps, eagerly between graph replays."""
        ext = _ops.require_ext()
        ext.per_rebuild(self.tree, self.capacity)

    def total(self) -> float:
        return float(self.tree[0])

    def __len__(self) -> int:
        return self.n_entries

    def state_dict(self) -> dict:
        return {"tree": self.tree, "data": self.data, "write": self.write,
                "n_entries": self.n_entries, "beta": self.beta}

    def load_state_dict(self, sd: dict) -> None:
        self.tree.copy_(sd["tree"])
        for k, v in sd["data"].items():
            self.data[k].copy_(v)
        self.write = sd["write"]
        self.n_entries = sd["n_entries"]
        self.beta = sd["beta"]
        self.beta_buf.fill_(self.beta)
        self.n_entries_buf.fill_(float(self.n_entries))
