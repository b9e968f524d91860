"""Race/stress test of the shm transport: 4 producer processes push
checksummed trajectories at full speed while the consumer drains; every
payload must arrive exactly once, uncorrupted (SURVEY §5.2 — the reference
shipped a torn-read race; this is the regression test for ours)."""

import multiprocessing as mp
import time

import numpy as np

from distributed_reinforcement_learning_amd.parallel.queue import (
    TrajectoryQueue,
)

SCHEMA = {
    "payload": ((64,), np.float32),
    "seq": ((1,), np.int64),
}
N_ACTORS = 4
PER_ACTOR = 200


def _producer(ns, task):
    rng = np.random.default_rng(task)
    q = TrajectoryQueue(SCHEMA, num_actors=N_ACTORS, queue_size=16,
                        role="actor", namespace=ns, actor_task=task)
    for k in range(PER_ACTOR):
        payload = rng.random(64).astype(np.float32)
        # checksum folded into slot 0: consumer recomputes over [1:]
        payload[0] = payload[1:].sum()
        q.append_to_queue(task, payload=payload,
                          seq=np.array([task * PER_ACTOR + k]))
    q.close()


def test_concurrent_producers_no_corruption():
    ns = f"st{int(time.time()*1000)%100000}"
    learner = TrajectoryQueue(SCHEMA, num_actors=N_ACTORS, queue_size=16,
                              role="learner", namespace=ns)
    procs = [mp.Process(target=_producer, args=(ns, t))
             for t in range(N_ACTORS)]
    try:
        for p in procs:
            p.start()
        seen = set()
        total = N_ACTORS * PER_ACTOR
        got = 0
        while got < total:
            batch = learner.sample_batch(min(32, total - got), timeout=60)
            n = batch["seq"].shape[0]
            for i in range(n):
                seq = int(batch["seq"][i, 0])
                assert seq not in seen, f"duplicate {seq}"
                seen.add(seq)
                payload = batch["payload"][i]
                np.testing.assert_allclose(payload[0], payload[1:].sum(),
                                           rtol=1e-5)
            got += n
        assert seen == set(range(total))
        for p in procs:
            p.join(timeout=15)
            assert p.exitcode == 0
    finally:
        for p in procs:
            if p.is_alive():
                p.terminate()
        learner.close()
