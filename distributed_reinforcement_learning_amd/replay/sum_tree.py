"""Array-backed sum tree with batched, iterative (vectorized) operations.

Capability-parity with reference buffer_queue.py:326-371 (same tree layout:
``tree[0 .. 2*cap-2]``, leaves at ``cap-1 ..``, ring-buffer writes), but
every op is batched and iterative:

* ``update_batch`` walks all touched leaves to the root level-by-level with
  numpy (no per-item Python recursion; duplicate parents handled via
  np.add.at on deltas),
* ``retrieve_batch`` descends all n queries in lock-step vectorized form.

The GPU twin (ops/hip/per_tree.hip) keeps the same layout so learner-side
priorities never round-trip to the host.
"""

from __future__ import annotations

import numpy as np


class SumTree:
    def __init__(self, capacity: int):
        self.capacity = int(capacity)
        self.tree = np.zeros(2 * self.capacity - 1, dtype=np.float64)
        self.data = np.empty(self.capacity, dtype=object)
        self.write = 0
        self.n_entries = 0

    def total(self) -> float:
        return float(self.tree[0])

    # -- single-item API (reference-compatible) -----------------------------

    def add(self, p: float, data) -> int:
        idx = self.write + self.capacity - 1
        self.data[self.write] = data
        self.update(idx, p)
        self.write = (self.write + 1) % self.capacity
        self.n_entries = min(self.n_entries + 1, self.capacity)
        return idx

    def update(self, idx: int, p: float) -> None:
        self.update_batch(np.asarray([idx]), np.asarray([p], dtype=np.float64))

    def get(self, s: float):
        idx = int(self.retrieve_batch(np.asarray([s], dtype=np.float64))[0])
        data_idx = idx - self.capacity + 1
        return idx, float(self.tree[idx]), self.data[data_idx]

    # -- batched API ---------------------------------------------------------

    def add_batch(self, ps: np.ndarray, datas) -> np.ndarray:
        n = len(ps)
        writes = (self.write + np.arange(n)) % self.capacity
        for w, d in zip(writes, datas):
            self.data[w] = d
        idxs = writes + self.capacity - 1
        self.update_batch(idxs, np.asarray(ps, dtype=np.float64))
        self.write = int((self.write + n) % self.capacity)
        self.n_entries = min(self.n_entries + n, self.capacity)
        return idxs

    def update_batch(self, idxs: np.ndarray, ps: np.ndarray) -> None:
        idxs = np.asarray(idxs, dtype=np.int64)
        # later duplicates win, like sequential reference updates
        last = {}
        for k, i in enumerate(idxs):
            last[int(i)] = k
        keep = np.fromiter(last.values(), dtype=np.int64)
        idxs = idxs[keep]
        ps = np.asarray(ps, dtype=np.float64)[keep]
        deltas = ps - self.tree[idxs]
        self.tree[idxs] = ps
        parents = (idxs - 1) // 2
        while True:
            np.add.at(self.tree, parents, deltas)
            root_mask = parents > 0
            if not root_mask.any():
                break
            parents = parents[root_mask]
            deltas = deltas[root_mask]
            parents = (parents - 1) // 2

    def retrieve_batch(self, s: np.ndarray) -> np.ndarray:
        """Vectorized root-to-leaf descent for n prefix-sum queries."""
        s = np.array(s, dtype=np.float64, copy=True)
        idx = np.zeros(len(s), dtype=np.int64)
        tree_len = len(self.tree)
        while True:
            left = 2 * idx + 1
            interior = left < tree_len
            if not interior.any():
                break
            li = left[interior]
            left_sums = self.tree[li]
            go_right = s[interior] > left_sums
            s[interior] = np.where(go_right, s[interior] - left_sums,
                                   s[interior])
            idx[interior] = np.where(go_right, li + 1, li)
        # rounding at a segment boundary can walk onto a zero-priority
        # never-written leaf (stale payload, infinite IS weight); clamp
        # into the populated range (ring writes fill the low slots first)
        last = self.capacity - 2 + max(self.n_entries, 1)
        return np.minimum(idx, last)

    def leaf_priorities(self, idxs: np.ndarray) -> np.ndarray:
        return self.tree[np.asarray(idxs, dtype=np.int64)]

    def state_dict(self) -> dict:
        return {"tree": self.tree.copy(), "data": self.data.copy(),
                "write": self.write, "n_entries": self.n_entries}

    def load_state_dict(self, sd: dict) -> None:
        self.tree = sd["tree"].copy()
        self.data = sd["data"].copy()
        self.write = sd["write"]
        self.n_entries = sd["n_entries"]
