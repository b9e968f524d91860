"""Failure detection / recovery / resume — subsystems the reference lacks
(SURVEY §5.3/§5.4)."""

import multiprocessing as mp
import os
import time

import numpy as np
import pytest


def _flaky_actor(task):
    # dies immediately; the supervisor must respawn it
    os._exit(1)


def test_supervisor_respawns_dead_actor():
    from distributed_reinforcement_learning_amd.parallel.heartbeat import (
        ActorSupervisor,
    )
    sup = ActorSupervisor(_flaky_actor, [0, 1], start_method="spawn")
    sup.start()
    time.sleep(3)
    respawned = sup.check()
    assert set(respawned) == {0, 1}
    assert sup.restarts[0] >= 1
    sup.stop()


def test_heartbeat_monitor_flags_stale():
    from distributed_reinforcement_learning_amd.parallel.heartbeat import (
        HeartbeatMonitor,
    )

    class FakeQueue:
        def heartbeats(self):
            return {0: time.time(), 1: time.time() - 1000.0}

    mon = HeartbeatMonitor(FakeQueue(), timeout=60.0)
    assert mon.stale_actors() == [1]


def test_checkpoint_resume_continues_training(tmp_path):
    """save_weights -> new agent -> load_weights -> training continues from
    the same step with identical weights and optimizer state."""
    from distributed_reinforcement_learning_amd.agents import a3c as a3c_agent
    import torch

    def make():
        return a3c_agent.Agent(
            input_shape=[4], num_action=2, discount_factor=0.99,
            start_learning_rate=1e-3, end_learning_rate=0.0,
            learning_frame=10 ** 6, baseline_loss_coef=1.0,
            entropy_coef=0.01, gradient_clip_norm=40.0,
            reward_clipping="none", seed=0)

    rng = np.random.default_rng(0)

    def batch(seed):
        r = np.random.default_rng(seed)
        N = 8
        return dict(state=r.normal(size=(N, 4)).astype(np.float32),
                    next_state=r.normal(size=(N, 4)).astype(np.float32),
                    previous_action=r.integers(0, 2, N),
                    action=r.integers(0, 2, N),
                    reward=r.normal(size=N).astype(np.float32),
                    done=np.zeros(N, dtype=bool))

    a1 = make()
    for i in range(3):
        a1.train(**batch(i))
    path = str(tmp_path / "ck.pt")
    a1.save_weights(path)
    # continue a1 two more steps
    for i in range(3, 5):
        a1.train(**batch(i))

    a2 = make()
    a2.load_weights(path)
    assert a2.global_step == 3
    for i in range(3, 5):
        a2.train(**batch(i))
    # identical continuation (same data, same optimizer state)
    assert torch.allclose(a1.optimizer.flat_params,
                          a2.optimizer.flat_params, atol=1e-6)
    assert a1.global_step == a2.global_step == 5
