"""Failure detection + actor supervision (absent in the reference —
SURVEY.md §5.3: a dead actor silently starves the queue; a dead learner hangs
every actor).

Actors already stamp a heartbeat into their ring header on every push
(parallel/queue.py). ``HeartbeatMonitor`` flags stale actors;
``ActorSupervisor`` respawns them (actors are stateless beyond env state, so a
respawn is cheap and sound).
"""

from __future__ import annotations

import multiprocessing as mp
import time
from typing import Callable, Dict, List, Optional


class HeartbeatMonitor:
    def __init__(self, queue, timeout: float = 60.0):
        self.queue = queue
        self.timeout = timeout

    def stale_actors(self) -> List[int]:
        now = time.time()
        return [i for i, hb in self.queue.heartbeats().items()
                if now - hb > self.timeout]


class ActorSupervisor:
    """Spawn + watch actor processes; restart any that die or go stale."""

    def __init__(self, actor_fn: Callable[[int], None], actor_ids: List[int],
                 monitor: Optional[HeartbeatMonitor] = None,
                 start_method: str = "spawn"):
        # spawn by default: a forked child of a learner that already ran
        # parallel torch ops deadlocks in the inherited OpenMP pool
        self.actor_fn = actor_fn
        self.actor_ids = list(actor_ids)
        self.monitor = monitor
        self._ctx = mp.get_context(start_method)
        self.procs: Dict[int, mp.process.BaseProcess] = {}
        self.restarts: Dict[int, int] = {i: 0 for i in self.actor_ids}

    def start(self) -> None:
        for i in self.actor_ids:
            self._spawn(i)

    def _spawn(self, actor_id: int) -> None:
        p = self._ctx.Process(target=self.actor_fn, args=(actor_id,),
                              daemon=True)
        p.start()
        self.procs[actor_id] = p

    def check(self) -> List[int]:
        """Respawn dead/stale actors; returns the list respawned."""
        respawned = []
        stale = set(self.monitor.stale_actors()) if self.monitor else set()
        for i, p in list(self.procs.items()):
            if not p.is_alive() or i in stale:
                if p.is_alive():
                    p.terminate()
                    p.join(timeout=5)
                self._spawn(i)
                self.restarts[i] += 1
                respawned.append(i)
        return respawned

    def stop(self) -> None:
        for p in self.procs.values():
            if p.is_alive():
                p.terminate()
        for p in self.procs.values():
            p.join(timeout=5)
