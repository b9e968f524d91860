"""Actor-side trajectory accumulators.

Mirrors the capability of reference utils.py:47-118
(``UnrolledA3CTrajectory`` / ``UnrolledTrajectory``): collect one unroll of
per-step fields, then hand the stacked arrays to the transport queue.
Implemented as a single generic accumulator parameterized by field names, with
the two reference-named aliases kept for API parity.
"""

from __future__ import annotations

from typing import Dict, List, Sequence

import numpy as np


class FieldTrajectory:
    """Accumulate per-step values for a fixed set of fields."""

    def __init__(self, fields: Sequence[str]):
        self.fields = tuple(fields)
        self._data: Dict[str, List] = {}
        self.initialize()

    def initialize(self) -> None:
        self._data = {f: [] for f in self.fields}

    def append(self, **kwargs) -> None:
        if set(kwargs) != set(self.fields):
            missing = set(self.fields) - set(kwargs)
            extra = set(kwargs) - set(self.fields)
            raise KeyError(f"trajectory fields mismatch: missing={missing} "
                           f"extra={extra}")
        for k, v in kwargs.items():
            self._data[k].append(v)

    def __len__(self) -> int:
        return len(self._data[self.fields[0]])

    def __getitem__(self, field: str) -> List:
        return self._data[field]

    def stacked(self) -> Dict[str, np.ndarray]:
        return {k: np.stack(v) for k, v in self._data.items()}


class UnrolledA3CTrajectory(FieldTrajectory):
    """A3C unroll: (s, s', pa, a, r, d) — reference utils.py:47-78."""

    def __init__(self):
        super().__init__(["state", "next_state", "previous_action",
                          "action", "reward", "done"])


class UnrolledTrajectory(FieldTrajectory):
    """IMPALA unroll incl. behavior policy and LSTM state —
    reference utils.py:80-118."""

    def __init__(self):
        super().__init__(["state", "next_state", "previous_action", "action",
                          "reward", "done", "behavior_policy",
                          "initial_h", "initial_c"])
