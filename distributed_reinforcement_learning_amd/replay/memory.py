"""Proportional prioritized replay (Schaul et al.) — capability-parity with
reference buffer_queue.py:373-416:

priority = (|err| + e)^a with e=0.001, a=0.6; stratified segment sampling;
IS weights (n * P(i))^-beta normalized by max, beta 0.4 -> 1.0 advancing
+0.001 per sample() call.
"""

from __future__ import annotations

import random
from typing import List, Tuple

import numpy as np

from distributed_reinforcement_learning_amd.replay.sum_tree import SumTree


class Memory:
    e = 0.001
    a = 0.6
    beta_start = 0.4
    beta_increment_per_sampling = 0.001

    def __init__(self, capacity: int, seed=None):
        self.capacity = int(capacity)
        self.tree = SumTree(self.capacity)
        self.beta = self.beta_start
        self.rng = np.random.default_rng(seed)

    def reset(self) -> None:
        self.tree = SumTree(self.capacity)
        self.beta = self.beta_start

    def _get_priority(self, error) -> np.ndarray:
        return (np.abs(np.asarray(error, dtype=np.float64)) + self.e) ** self.a

    def add(self, error: float, sample) -> None:
        self.tree.add(float(self._get_priority(error)), sample)

    def add_batch(self, errors: np.ndarray, samples) -> None:
        self.tree.add_batch(self._get_priority(errors), samples)

    def sample(self, n: int) -> Tuple[List, np.ndarray, np.ndarray]:
        total = self.tree.total()
        segment = total / n
        self.beta = min(1.0, self.beta + self.beta_increment_per_sampling)
        lo = segment * np.arange(n)
        s = lo + self.rng.random(n) * segment
        idxs = self.tree.retrieve_batch(s)
        priorities = self.tree.leaf_priorities(idxs)
        data_idxs = idxs - self.tree.capacity + 1
        batch = [self.tree.data[i] for i in data_idxs]
        probs = priorities / total
        is_weight = np.power(self.tree.n_entries * probs, -self.beta)
        is_weight /= is_weight.max()
        return batch, idxs, is_weight.astype(np.float32)

    def update(self, idx: int, error: float) -> None:
        self.tree.update(int(idx), float(self._get_priority(error)))

    def update_batch(self, idxs: np.ndarray, errors: np.ndarray) -> None:
        self.tree.update_batch(np.asarray(idxs), self._get_priority(errors))

    def __len__(self) -> int:
        return self.tree.n_entries

    def state_dict(self) -> dict:
        return {"tree": self.tree.state_dict(), "beta": self.beta}

    def load_state_dict(self, sd: dict) -> None:
        self.tree.load_state_dict(sd["tree"])
        self.beta = sd["beta"]
