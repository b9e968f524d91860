"""Actor-loop throughput: scalar (reference topology, batch-1 inference)
vs vectorized (envs_per_actor=E, one batched forward per tick).

CPU-only — actors are CPU processes by design (BASELINE topology). This
measures ONE actor process; the full-system rate is ~num_actors x this
(each process is independent). VERDICT r1 item 8 target: >=10k
env-frames/s end-to-end with 20 processes, i.e. >=500 frames/s/process.
"""
import json
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

_THREADS = int(os.environ.get("ACTOR_THREADS", "1"))
torch.set_num_threads(_THREADS)
try:
    import threadpoolctl
    _TP = threadpoolctl.threadpool_limits(_THREADS)
except Exception:
    pass

from distributed_reinforcement_learning_amd.agents import impala
from distributed_reinforcement_learning_amd.envs import make_uint8_env


def build_agent():
    return impala.Agent(
        trajectory=20, input_shape=[84, 84, 4], num_action=18,
        lstm_hidden_size=256, discount_factor=0.99,
        start_learning_rate=6e-4, end_learning_rate=0.0,
        learning_frame=10 ** 9, baseline_loss_coef=1.0, entropy_coef=0.05,
        gradient_clip_norm=40.0, reward_clipping="abs_one", device="cpu",
        build_optimizer=False, seed=1)


def bench_scalar(ticks=120):
    agent = build_agent()
    env = make_uint8_env("BreakoutDeterministic-v4", num_actions=18, seed=0)
    state = env.reset()
    pa, h, c = 0, np.zeros(256, np.float32), np.zeros(256, np.float32)
    t0 = time.perf_counter()
    for _ in range(ticks):
        a, pol, mp, h, c = agent.get_policy_and_action(state, pa, h, c)
        state, r, d, info = env.step(a % 4)
        pa = a
        if d:
            state = env.reset()
    dt = time.perf_counter() - t0
    return ticks / dt


def bench_vector(E, ticks=120):
    agent = build_agent()
    envs = [make_uint8_env("BreakoutDeterministic-v4", num_actions=18,
                           seed=e) for e in range(E)]
    states = np.stack([e.reset() for e in envs])
    pa = np.zeros(E, np.int64)
    h = np.zeros((E, 256), np.float32)
    c = np.zeros((E, 256), np.float32)
    t0 = time.perf_counter()
    for _ in range(ticks):
        actions, pol, mp, h, c = agent.get_policy_and_action_batch(
            states, pa, h, c)
        for e, env in enumerate(envs):
            s, r, d, info = env.step(int(actions[e]) % 4)
            states[e] = s if not d else env.reset()
        pa = actions
    dt = time.perf_counter() - t0
    return E * ticks / dt


if __name__ == "__main__":
    out = {"threads": torch.get_num_threads()}
    if os.environ.get("SKIP_SCALAR") != "1":
        out["scalar_frames_per_s"] = round(bench_scalar(), 1)
    for E in (16, 32, 64):
        out[f"vector{E}_frames_per_s"] = round(bench_vector(E), 1)
    print(json.dumps(out), flush=True)
