"""Per-stage wall-clock timing for the learner hot loop.

The reference only logs a single per-train-step delta (train_impala.py:113).
Here every stage (ingest / sample / H2D / forward-backward / all-reduce /
optimizer) gets its own exponential-moving-average timer so regressions are
attributable, and the aggregate feeds the env-frames/sec headline metric.
"""

from __future__ import annotations

import time
from typing import Dict


class StageTimer:
    def __init__(self, ema: float = 0.98):
        self._ema = ema
        self._avg: Dict[str, float] = {}
        self._count: Dict[str, int] = {}
        self._t0: Dict[str, float] = {}

    def start(self, stage: str) -> None:
        self._t0[stage] = time.perf_counter()

    def stop(self, stage: str) -> float:
        dt = time.perf_counter() - self._t0.pop(stage)
        if stage in self._avg:
            self._avg[stage] = self._ema * self._avg[stage] + (1 - self._ema) * dt
        else:
            self._avg[stage] = dt
        self._count[stage] = self._count.get(stage, 0) + 1
        return dt

    class _Ctx:
        def __init__(self, timer: "StageTimer", stage: str):
            self.timer, self.stage = timer, stage

        def __enter__(self):
            self.timer.start(self.stage)

        def __exit__(self, *exc):
            self.timer.stop(self.stage)

    def track(self, stage: str) -> "StageTimer._Ctx":
        return StageTimer._Ctx(self, stage)

    def averages(self) -> Dict[str, float]:
        return dict(self._avg)

    def report(self) -> str:
        return " ".join(f"{k}={v*1e3:.2f}ms" for k, v in
                        sorted(self._avg.items()))
