#!/usr/bin/env python3
"""GPU triage: (1) hipGraph replay overhead microbench, (2) GraphedImpalaStep
stage timings, (3) smoke teardown bisect helpers."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def micro_graph():
    """Time replay of a graph of K tiny kernels vs K eager launches."""
    x = torch.ones(1024, device="cuda")
    K = 300
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        for _ in range(3):
            y = x
            for _ in range(K):
                y = y * 1.0001
    torch.cuda.current_stream().wait_stream(s)
    g = torch.cuda.CUDAGraph()
    with torch.cuda.graph(g):
        y = x
        for _ in range(K):
            y = y * 1.0001
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(50):
        g.replay()
    torch.cuda.synchronize()
    t_graph = (time.perf_counter() - t0) / 50
    t0 = time.perf_counter()
    for _ in range(50):
        y = x
        for _ in range(K):
            y = y * 1.0001
    torch.cuda.synchronize()
    t_eager = (time.perf_counter() - t0) / 50
    print(f"[micro] {K} kernels: graph replay {t_graph*1e3:.3f} ms, "
          f"eager {t_eager*1e3:.3f} ms", flush=True)


def staged_impala():
    from distributed_reinforcement_learning_amd.agents import impala
    from distributed_reinforcement_learning_amd.runtime import GraphedImpalaStep

    B, T, A, H = 32, 20, 18, 256
    agent = impala.Agent(
        trajectory=T, input_shape=[84, 84, 4], num_action=A,
        lstm_hidden_size=H, discount_factor=0.99, start_learning_rate=6e-4,
        end_learning_rate=0.0, learning_frame=10 ** 9,
        baseline_loss_coef=1.0, entropy_coef=0.05, gradient_clip_norm=40.0,
        reward_clipping="abs_one", device="cuda:0", seed=0)
    gs = GraphedImpalaStep(agent, B)
    rng = np.random.default_rng(0)
    batch = dict(
        state=rng.integers(0, 255, (B, T, 84, 84, 4), dtype=np.uint8),
        reward=rng.normal(size=(B, T)).astype(np.float32),
        action=rng.integers(0, A, (B, T)).astype(np.int32),
        done=np.zeros((B, T), dtype=bool),
        behavior_policy=np.full((B, T, A), 1 / A, dtype=np.float32),
        previous_action=rng.integers(0, A, (B, T)).astype(np.int32),
        initial_h=np.zeros((B, T, H), dtype=np.float32),
        initial_c=np.zeros((B, T, H), dtype=np.float32),
    )
    # warm
    for _ in range(5):
        gs.step(batch)
    torch.cuda.synchronize()

    def tsec(fn, n=20, sync=True):
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        for _ in range(n):
            fn()
        if sync:
            torch.cuda.synchronize()
        return (time.perf_counter() - t0) / n * 1e3

    print(f"[staged] load_inputs  {tsec(lambda: gs.load_inputs(batch)):.3f} ms")
    print(f"[staged] fill lr      {tsec(lambda: gs.lr_buf.fill_(1e-4)):.3f} ms")
    print(f"[staged] g_fwd_bwd    {tsec(gs.g_fwd_bwd.replay):.3f} ms")
    print(f"[staged] g_opt        {tsec(gs.g_opt.replay):.3f} ms")
    print(f"[staged] losses->host "
          f"{tsec(lambda: [float(x) for x in gs.losses]):.3f} ms")

    def combo_load_replay():
        gs.load_inputs(batch)
        gs.g_fwd_bwd.replay()

    def combo_replay_replay():
        gs.g_fwd_bwd.replay()
        gs.g_opt.replay()

    def combo_load_replay_replay():
        gs.load_inputs(batch)
        gs.g_fwd_bwd.replay()
        gs.g_opt.replay()

    def combo_lr_replay():
        gs.lr_buf.fill_(1e-4)
        gs.g_fwd_bwd.replay()

    def step_no_losses():
        gs.load_inputs(batch)
        gs.lr_buf.fill_(1e-4)
        gs.g_fwd_bwd.replay()
        gs.g_opt.replay()

    print(f"[staged] load+fb      {tsec(combo_load_replay):.3f} ms")
    print(f"[staged] fb+opt       {tsec(combo_replay_replay):.3f} ms")
    print(f"[staged] lr+fb        {tsec(combo_lr_replay):.3f} ms")
    print(f"[staged] load+fb+opt  {tsec(combo_load_replay_replay):.3f} ms")
    print(f"[staged] step-no-loss {tsec(step_no_losses):.3f} ms")
    print(f"[staged] full step    {tsec(lambda: gs.step(batch)):.3f} ms")
    # eager comparison
    t = tsec(lambda: agent.train(
        state=batch['state'], reward=batch['reward'], action=batch['action'],
        done=batch['done'], behavior_policy=batch['behavior_policy'],
        previous_action=batch['previous_action'],
        initial_h=batch['initial_h'], initial_c=batch['initial_c']), n=10)
    print(f"[staged] eager train  {t:.3f} ms", flush=True)


if __name__ == "__main__":
    which = sys.argv[1] if len(sys.argv) > 1 else "all"
    if which in ("micro", "all"):
        micro_graph()
    if which in ("staged", "all"):
        staged_impala()
