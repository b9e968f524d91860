"""Actor-side uniform replay (reference buffer_queue.py:283-324)."""

from __future__ import annotations

import collections
from typing import Dict, List

import numpy as np


class LocalBuffer:
    FIELDS = ("state", "next_state", "previous_action", "action",
              "reward", "done")

    def __init__(self, capacity: int, seed=None):
        self._d: Dict[str, collections.deque] = {
            f: collections.deque(maxlen=int(capacity)) for f in self.FIELDS}
        self.rng = np.random.default_rng(seed)

    def append(self, state, next_state, previous_action, action,
               reward, done) -> None:
        vals = dict(state=state, next_state=next_state,
                    previous_action=previous_action, action=action,
                    reward=reward, done=done)
        for k, v in vals.items():
            self._d[k].append(v)

    def sample(self, batch_size: int) -> Dict[str, List]:
        n = len(self)
        idxs = self.rng.permutation(n)[:batch_size]
        return {f: [self._d[f][i] for i in idxs] for f in self.FIELDS}

    def __len__(self) -> int:
        return len(self._d["state"])
