"""Evaluate a trained checkpoint: run N episodes with the actor policy and
report mean/max score + episode length (the reference only exposes scores
through the actors' TensorBoard scalars; this is the standalone
counterpart).

    python scripts/evaluate.py --algorithm impala --checkpoint \\
        checkpoints/impala.pt [--episodes 10] [--config config.json]
        [--algorithm_block impala] [--greedy]
"""

import argparse
import json
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

torch.set_num_threads(1)


def build(algorithm: str, cfg, device: str):
    if algorithm == "impala":
        from distributed_reinforcement_learning_amd.trainers import impala as t
    elif algorithm == "a3c":
        from distributed_reinforcement_learning_amd.trainers import a3c as t
    elif algorithm == "apex":
        from distributed_reinforcement_learning_amd.trainers import apex as t
    elif algorithm == "r2d2":
        from distributed_reinforcement_learning_amd.trainers import r2d2 as t
    else:
        raise SystemExit(f"unknown algorithm {algorithm}")

    class _Ctx:
        pass

    ctx = _Ctx()
    ctx.cfg = cfg
    if algorithm in ("impala",):
        return t.build_agent(ctx, device, build_optimizer=False, seed=0)
    return t.build_agent(ctx, device, False, 0)


def run_episode(algorithm, agent, env, cfg, greedy: bool):
    state = env.reset()
    prev_action = 0
    H = int(cfg.get("lstm_size", 0) or 0)
    h = np.zeros(H, np.float32) if H else None
    c = np.zeros(H, np.float32) if H else None
    score, steps = 0.0, 0
    avail = cfg.available_action[0]
    while True:
        if algorithm == "impala":
            a, policy, _, h, c = agent.get_policy_and_action(
                state, prev_action, h, c)
            if greedy:
                a = int(np.argmax(policy))
        elif algorithm == "a3c":
            a, policy, _ = agent.get_policy_and_action(state, prev_action)
            if greedy:
                a = int(np.argmax(policy))
        elif algorithm == "apex":
            a, _, _ = agent.get_policy_and_action(
                state, prev_action, 0.0 if greedy else 0.05)
        else:  # r2d2
            a, _, h, c = agent.get_action(state, h, c, prev_action,
                                          0.0 if greedy else 0.05)
        env_action = a % avail if algorithm != "r2d2" else a
        state, r, done, info = env.step(env_action)
        prev_action = a
        score += r
        steps += 1
        if done or steps >= 100_000:
            return score, steps


def main(argv=None):
    from distributed_reinforcement_learning_amd.config import (
        default_config_path, load_config,
    )
    from distributed_reinforcement_learning_amd.envs import (
        make_env, make_uint8_env, pomdp_uint8_env,
    )

    p = argparse.ArgumentParser()
    p.add_argument("--algorithm", required=True,
                   choices=["a3c", "impala", "apex", "r2d2"])
    p.add_argument("--checkpoint", required=True)
    p.add_argument("--episodes", type=int, default=10)
    p.add_argument("--config", default=default_config_path())
    p.add_argument("--algorithm_block", default=None)
    p.add_argument("--greedy", action="store_true",
                   help="argmax policy / epsilon=0 instead of sampling")
    p.add_argument("--seed", type=int, default=0)
    args = p.parse_args(argv)

    cfg = load_config(args.config, args.algorithm_block or args.algorithm)
    agent = build(args.algorithm, cfg, "cpu")
    agent.load_weights(args.checkpoint)

    env_name = cfg.env[0]
    if args.algorithm == "r2d2":
        env = pomdp_uint8_env(env_name, num_actions=cfg.model_output,
                              seed=args.seed)
    elif env_name.startswith("CartPole"):
        env = make_env(env_name, seed=args.seed)
    else:
        env = make_uint8_env(env_name, num_actions=cfg.model_output,
                             seed=args.seed)

    scores, lengths = [], []
    for ep in range(args.episodes):
        s, n = run_episode(args.algorithm, agent, env, cfg, args.greedy)
        scores.append(s)
        lengths.append(n)
        print(f"episode {ep}: score={s:.1f} steps={n}", flush=True)
    print(json.dumps({
        "algorithm": args.algorithm,
        "checkpoint": args.checkpoint,
        "episodes": args.episodes,
        "mean_score": float(np.mean(scores)),
        "max_score": float(np.max(scores)),
        "mean_episode_steps": float(np.mean(lengths)),
        "greedy": bool(args.greedy),
    }), flush=True)


if __name__ == "__main__":
    main()
