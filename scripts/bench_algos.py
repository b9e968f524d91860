"""DEPRECATED in favor of ``python bench.py --algo {apex,r2d2}`` (the
driver-contract harness with warmup/clock-priming and the standard JSON
line); kept for r1 comparability.

Per-algorithm GPU learner microbenches (Ape-X and R2D2): synthetic
replay contents, device-resident PER shard, reference configs. Reports
train-steps/s and transitions/s per algorithm (one JSON line each).

The driver's headline bench (bench.py) is IMPALA; these numbers back the
BASELINE.md per-algorithm table.
"""
import json
import sys
import os
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from distributed_reinforcement_learning_amd.config import load_config
from distributed_reinforcement_learning_amd.replay.gpu_memory import GpuMemory


def bench_apex(steps=200, warmup=30):
    from distributed_reinforcement_learning_amd.agents import apex as apex_agent
    cfg = load_config("config.json", "apex")
    dev = "cuda:0"
    agent = apex_agent.Agent(
        input_shape=cfg.model_input, num_action=cfg.model_output,
        discount_factor=cfg.discount_factor,
        gradient_clip_norm=cfg.gradient_clip_norm,
        reward_clipping=cfg.reward_clipping,
        start_learning_rate=cfg.start_learning_rate,
        end_learning_rate=cfg.end_learning_rate,
        learning_frame=cfg.learning_frame, device=dev,
        build_optimizer=True, seed=3)
    H, W, C = cfg.model_input
    mem = GpuMemory(100_000, fields={
        "state": ((H, W, C), torch.uint8),
        "next_state": ((H, W, C), torch.uint8),
        "previous_action": ((), torch.int32),
        "action": ((), torch.int32),
        "reward": ((), torch.float32),
        "done": ((), torch.bool)}, device=dev, seed=3)
    rng = np.random.default_rng(0)
    T = cfg.trajectory
    for _ in range(40):
        dev_u = {
            "state": torch.as_tensor(rng.integers(0, 255, (T, H, W, C),
                                                  dtype=np.uint8)).to(dev),
            "next_state": torch.as_tensor(
                rng.integers(0, 255, (T, H, W, C), dtype=np.uint8)).to(dev),
            "previous_action": torch.as_tensor(
                rng.integers(0, cfg.model_output, T).astype(np.int32)).to(dev),
            "action": torch.as_tensor(
                rng.integers(0, cfg.model_output, T).astype(np.int32)).to(dev),
            "reward": torch.as_tensor(
                rng.normal(size=T).astype(np.float32)).to(dev),
            "done": torch.as_tensor(rng.random(T) < 0.02).to(dev)}
        td = agent.get_td_error(
            dev_u["state"], dev_u["next_state"], dev_u["previous_action"],
            dev_u["action"], dev_u["reward"], dev_u["done"], as_tensor=True)
        mem.add_batch(td, dev_u)

    from distributed_reinforcement_learning_amd.runtime import (
        GraphedReplayStep,
    )

    def loss_fn(b, w):
        return agent.compute_distributed_loss(
            agent.frames_to_device(b["state"]),
            agent.frames_to_device(b["next_state"]),
            b["previous_action"].long(), b["action"].long(),
            b["reward"], b["done"], w)

    graphed = GraphedReplayStep(agent, mem, cfg.batch_size, loss_fn)

    def one():
        graphed.step()

    for _ in range(warmup):
        one()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        one()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    print(json.dumps({
        "algo": "apex", "ms_per_train_step": dt * 1000,
        "train_steps_per_s": 1 / dt,
        "transitions_per_s": cfg.batch_size / dt,
        "batch": cfg.batch_size}), flush=True)


def bench_r2d2(steps=200, warmup=30):
    from distributed_reinforcement_learning_amd.agents import r2d2 as r2d2_agent
    cfg = load_config("config.json", "r2d2")
    dev = "cuda:0"
    agent = r2d2_agent.Agent(
        seq_len=cfg.seq_len, burn_in=cfg.burn_in,
        input_shape=cfg.model_input, num_action=cfg.model_output,
        lstm_size=cfg.lstm_size, discount_factor=cfg.discount_factor,
        start_learning_rate=cfg.start_learning_rate,
        end_learning_rate=cfg.end_learning_rate,
        learning_frame=cfg.learning_frame,
        gradient_clip_norm=cfg.gradient_clip_norm,
        reward_clipping=cfg.reward_clipping, device=dev,
        build_optimizer=True, seed=3)
    H, W, C = cfg.model_input
    L = cfg.seq_len
    mem = GpuMemory(100_000, fields={
        "state": ((L, H, W, C), torch.uint8),
        "previous_action": ((L,), torch.int32),
        "action": ((L,), torch.int32),
        "reward": ((L,), torch.float32),
        "done": ((L,), torch.bool),
        "initial_h": ((L, cfg.lstm_size), torch.float32),
        "initial_c": ((L, cfg.lstm_size), torch.float32)},
        device=dev, seed=3)
    rng = np.random.default_rng(0)
    for _ in range(40):
        B = 4
        dev_u = {
            "state": torch.as_tensor(rng.integers(
                0, 255, (B, L, H, W, C), dtype=np.uint8)).to(dev),
            "previous_action": torch.as_tensor(rng.integers(
                0, cfg.model_output, (B, L)).astype(np.int32)).to(dev),
            "action": torch.as_tensor(rng.integers(
                0, cfg.model_output, (B, L)).astype(np.int32)).to(dev),
            "reward": torch.as_tensor(
                rng.normal(size=(B, L)).astype(np.float32)).to(dev),
            "done": torch.as_tensor(rng.random((B, L)) < 0.02).to(dev),
            "initial_h": torch.zeros(B, L, cfg.lstm_size, device=dev),
            "initial_c": torch.zeros(B, L, cfg.lstm_size, device=dev)}
        td = agent.get_td_error_batch(
            dev_u["state"], dev_u["previous_action"], dev_u["action"],
            dev_u["initial_h"][:, 0], dev_u["initial_c"][:, 0],
            dev_u["reward"], dev_u["done"], as_tensor=True)
        mem.add_batch(td, dev_u)

    from distributed_reinforcement_learning_amd.runtime import (
        GraphedReplayStep,
    )

    def loss_fn(b, w):
        # compute_sequence_loss preps device tensors internally
        return agent.compute_sequence_loss(
            b["state"], b["previous_action"], b["action"],
            b["initial_h"][:, 0], b["initial_c"][:, 0], b["reward"],
            b["done"], w)

    graphed = GraphedReplayStep(agent, mem, cfg.batch_size, loss_fn)

    def one():
        graphed.step()

    for _ in range(warmup):
        one()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(steps):
        one()
    torch.cuda.synchronize()
    dt = (time.perf_counter() - t0) / steps
    print(json.dumps({
        "algo": "r2d2", "ms_per_train_step": dt * 1000,
        "train_steps_per_s": 1 / dt,
        "sequences_per_s": cfg.batch_size / dt,
        "batch": cfg.batch_size, "seq_len": L, "burn_in": cfg.burn_in}),
        flush=True)


if __name__ == "__main__":
    from distributed_reinforcement_learning_amd.utils import tunableop
    tunableop.enable()
    bench_apex()
    bench_r2d2()
