"""Shared-memory trajectory queue + weight publication, including real
multi-process producer/consumer runs."""

import multiprocessing as mp
import time

import numpy as np
import pytest
import torch

from distributed_reinforcement_learning_amd.parallel.queue import (
    TrajectoryQueue, TrajectoryRing,
)
from distributed_reinforcement_learning_amd.parallel.weights import (
    WeightPublisher, WeightSubscriber,
)

SCHEMA = {
    "state": ((4, 8, 8, 2), np.uint8),
    "action": ((4,), np.int32),
    "reward": ((4,), np.float32),
    "done": ((4,), np.bool_),
}


def _mk_fields(k):
    return {
        "state": np.full((4, 8, 8, 2), k % 255, dtype=np.uint8),
        "action": np.full((4,), k, dtype=np.int32),
        "reward": np.full((4,), float(k), dtype=np.float32),
        "done": np.zeros((4,), dtype=np.bool_),
    }


def test_ring_push_pop_roundtrip():
    ring = TrajectoryRing("drla_test_ring0", SCHEMA, capacity=4, create=True)
    try:
        assert ring.size() == 0
        assert ring.try_push(_mk_fields(3))
        assert ring.size() == 1
        out = {n: np.empty((1, *s), d) for n, (s, d) in SCHEMA.items()}
        assert ring.try_pop_into(out, 0)
        assert out["action"][0, 0] == 3
        assert out["reward"][0, 2] == 3.0
        assert ring.size() == 0
        assert not ring.try_pop_into(out, 0)
    finally:
        ring.close()


def test_ring_fills_and_rejects():
    ring = TrajectoryRing("drla_test_ring1", SCHEMA, capacity=2, create=True)
    try:
        assert ring.try_push(_mk_fields(0))
        assert ring.try_push(_mk_fields(1))
        assert not ring.try_push(_mk_fields(2))
        assert not ring.push(_mk_fields(2), block=False)
    finally:
        ring.close()


def _producer(ns, n_items, task):
    q = TrajectoryQueue(SCHEMA, num_actors=2, queue_size=8, role="actor",
                        namespace=ns, actor_task=task)
    for k in range(n_items):
        q.append_to_queue(task, **_mk_fields(k * 2 + task))


def test_queue_multiprocess_scatter():
    ns = f"t{int(time.time()*1000)%100000}"
    learner = TrajectoryQueue(SCHEMA, num_actors=2, queue_size=8,
                              role="learner", namespace=ns)
    try:
        procs = [mp.Process(target=_producer, args=(ns, 6, t))
                 for t in range(2)]
        for p in procs:
            p.start()
        batch = learner.sample_batch(12, timeout=30)
        assert batch["state"].shape == (12, 4, 8, 8, 2)
        assert batch["state"].dtype == np.uint8
        # all 12 distinct payloads arrive exactly once
        assert sorted(batch["action"][:, 0].tolist()) == list(range(12))
        for p in procs:
            p.join(timeout=10)
            assert p.exitcode == 0
        assert learner.get_size() == 0
    finally:
        learner.close()


def test_queue_rank_sharding():
    ns = f"s{int(time.time()*1000)%100000}"
    l0 = TrajectoryQueue(SCHEMA, num_actors=4, queue_size=8, role="learner",
                         namespace=ns, rank=0, world_size=2)
    l1 = TrajectoryQueue(SCHEMA, num_actors=4, queue_size=8, role="learner",
                         namespace=ns, rank=1, world_size=2)
    try:
        assert l0.actor_ids == [0, 2]
        assert l1.actor_ids == [1, 3]
    finally:
        l0.close()
        l1.close()


def _subscriber_proc(name, q):
    model = torch.nn.Linear(10, 3)
    sub = WeightSubscriber(name, model.state_dict())
    sub.wait_for_first(timeout=30)
    sd = model.state_dict()
    step = sub.pull(sd)
    model.load_state_dict(sd)
    q.put((step, model.weight.detach().sum().item()))
    sub.close()


def test_weight_publish_subscribe_across_processes():
    torch.manual_seed(0)
    model = torch.nn.Linear(10, 3)
    name = f"drla_w{int(time.time()*1000)%100000}"
    pub = WeightPublisher(name, model.state_dict())
    try:
        q = mp.Queue()
        p = mp.Process(target=_subscriber_proc, args=(name, q))
        p.start()
        pub.publish(model.state_dict(), global_step=42)
        step, wsum = q.get(timeout=30)
        p.join(timeout=10)
        assert step == 42
        assert abs(wsum - model.weight.sum().item()) < 1e-5
    finally:
        pub.close()


def test_weight_version_skips_stale_pull():
    model = torch.nn.Linear(4, 2)
    name = f"drla_v{int(time.time()*1000)%100000}"
    pub = WeightPublisher(name, model.state_dict())
    try:
        sub = WeightSubscriber(name, model.state_dict())
        assert sub.pull(model.state_dict()) is None  # nothing published
        pub.publish(model.state_dict(), global_step=7)
        assert sub.pull(model.state_dict()) == 7
        assert sub.pull(model.state_dict()) is None  # no new version
        pub.publish(model.state_dict(), global_step=8)
        assert sub.pull(model.state_dict()) == 8
        sub.close()
    finally:
        pub.close()


def test_ring_wraparound_many_cycles():
    """Sequence numbers and slot reuse stay consistent across many full
    wraps of a small ring (the index arithmetic is modular)."""
    ring = TrajectoryRing("drla_test_wrap", SCHEMA, capacity=3, create=True)
    try:
        out = {n: np.empty((1, *s), d) for n, (s, d) in SCHEMA.items()}
        k = 0
        popped = []
        for cycle in range(50):
            while ring.try_push(_mk_fields(k % 255)):
                k += 1
            assert ring.size() == 3
            while ring.try_pop_into(out, 0):
                popped.append(int(out["action"][0, 0]))
        # FIFO order preserved across every wrap
        assert popped == [i % 255 for i in range(len(popped))]
        assert len(popped) == k
    finally:
        ring.close()


def test_weight_publisher_many_updates_never_torn():
    """Rapid republish with a concurrent reader: every successful pull is
    internally consistent (all entries carry the same fill value)."""
    sd = {"a": torch.zeros(64), "b": torch.zeros(32, 3)}
    pub = WeightPublisher("drla_test_churn", sd)
    try:
        sub = WeightSubscriber("drla_test_churn", {
            "a": torch.empty(64), "b": torch.empty(32, 3)})
        dest = {"a": torch.empty(64), "b": torch.empty(32, 3)}
        last = -1
        for step in range(200):
            v = float(step)
            pub.publish({"a": torch.full((64,), v),
                         "b": torch.full((32, 3), v)}, global_step=step)
            got = sub.pull(dest)
            if got is not None:
                a0 = float(dest["a"][0])
                assert torch.all(dest["a"] == a0)
                assert torch.all(dest["b"] == a0)
                assert got > last
                last = got
        assert last >= 0
    finally:
        pub.close()
