"""Stability soak: many thousands of graphed learner steps per algorithm
with device-memory and throughput drift checks (leaks or allocator growth
inside the replayed graphs would show up as rising reserved bytes; a
degrading step time as a rising tail rate)."""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distributed_reinforcement_learning_amd.utils import tunableop
tunableop.enable()

import numpy as np
import torch


def mem():
    return (torch.cuda.memory_allocated() / 2 ** 20,
            torch.cuda.memory_reserved() / 2 ** 20)


def soak_impala(steps=20000):
    import bench
    from distributed_reinforcement_learning_amd.agents import impala
    from distributed_reinforcement_learning_amd.runtime import (
        GraphedImpalaStep,
    )
    agent = impala.Agent(
        trajectory=20, input_shape=[84, 84, 4], num_action=18,
        lstm_hidden_size=256, discount_factor=0.99,
        start_learning_rate=6e-4, end_learning_rate=0.0,
        learning_frame=10 ** 9, baseline_loss_coef=1.0, entropy_coef=0.05,
        gradient_clip_norm=40.0, reward_clipping="abs_one",
        device="cuda:0", seed=1)
    g = GraphedImpalaStep(agent, 32)
    rng = np.random.default_rng(0)
    B, T, A, H = 32, 20, 18, 256
    batch = dict(
        state=rng.integers(0, 255, (B, T, 84, 84, 4), dtype=np.uint8),
        reward=rng.normal(size=(B, T)).astype(np.float32),
        action=rng.integers(0, A, (B, T)).astype(np.int32),
        done=(rng.random((B, T)) < 0.02),
        behavior_policy=np.full((B, T, A), 1.0 / A, dtype=np.float32),
        previous_action=rng.integers(0, A, (B, T)).astype(np.int32),
        initial_h=(rng.normal(size=(B, T, H)) * 0.1).astype(np.float32),
        initial_c=(rng.normal(size=(B, T, H)) * 0.1).astype(np.float32))
    g.stage_to_pinned(batch)
    for _ in range(50):
        g.step()
    torch.cuda.synchronize()
    a0, r0 = mem()
    t0 = time.perf_counter()
    for i in range(steps):
        g.step()
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    a1, r1 = mem()
    print(json.dumps({
        "algo": "impala", "steps": steps, "ms_per_step": dt / steps * 1e3,
        "alloc_mb": [round(a0, 1), round(a1, 1)],
        "reserved_mb": [round(r0, 1), round(r1, 1)],
        "losses_finite": all(np.isfinite(g.last_losses())),
    }), flush=True)
    assert abs(a1 - a0) < 64 and abs(r1 - r0) < 64, "device memory drift"


def soak_replay(algo, steps=10000):
    from bench import bench_apex, bench_r2d2  # noqa: F401  (reuse setup)
    import bench as bench_mod

    class Args:
        batch = None
        seq_len = 15
        burn_in = 7
        steps = 0
        warmup = 0
        min_warm_s = 0.0
        step_times = False

    # build through the bench helpers by monkeypatching run_timed
    holder = {}

    def fake_run_timed(step, args, world, have_gpu, device):
        holder["step"] = step
        return 1.0

    orig = bench_mod.run_timed
    bench_mod.run_timed = fake_run_timed
    orig_emit = bench_mod.emit
    bench_mod.emit = lambda *a, **k: None
    try:
        if algo == "apex":
            bench_mod.bench_apex(Args(), 0, 1, 0, True, "cuda:0")
        else:
            bench_mod.bench_r2d2(Args(), 0, 1, 0, True, "cuda:0")
    finally:
        bench_mod.run_timed = orig
        bench_mod.emit = orig_emit
    step = holder["step"]
    for i in range(50):
        step(i)
    torch.cuda.synchronize()
    a0, r0 = mem()
    t0 = time.perf_counter()
    for i in range(steps):
        step(i)
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    a1, r1 = mem()
    print(json.dumps({
        "algo": algo, "steps": steps, "ms_per_step": dt / steps * 1e3,
        "alloc_mb": [round(a0, 1), round(a1, 1)],
        "reserved_mb": [round(r0, 1), round(r1, 1)],
    }), flush=True)
    assert abs(a1 - a0) < 64 and abs(r1 - r0) < 64, "device memory drift"


if __name__ == "__main__":
    soak_impala()
    torch.cuda.empty_cache()
    soak_replay("apex")
    torch.cuda.empty_cache()
    soak_replay("r2d2")
    print("SOAK_OK", flush=True)
