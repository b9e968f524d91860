"""IMPALA trainer loops (reference train_impala.py).

Learner (reference :89-113): drain the trajectory rings, one batched V-trace
update per batch, publish weights, log the reference's scalar names
(data/pi_loss, data/value_loss, data/entropy, data/lr, data/time —
train_impala.py:109-113) plus per-stage timings and env-frames/sec.

Actor (reference :115-194): T-step unroll loop with stored per-step LSTM
state, behavior policy, life-loss shaping, weight pull once per unroll.
"""

from __future__ import annotations

import time

import numpy as np
import torch

from distributed_reinforcement_learning_amd.agents import impala as impala_agent
from distributed_reinforcement_learning_amd.envs import make_uint8_env
from distributed_reinforcement_learning_amd.parallel.queue import (
    TrajectoryQueue, queue_schema_for,
)
from distributed_reinforcement_learning_amd.parallel.weights import (
    WeightPublisher, WeightSubscriber,
)
from distributed_reinforcement_learning_amd.trainers import common
from distributed_reinforcement_learning_amd.utils import (
    StageTimer, SummaryWriter, UnrolledTrajectory,
)


def build_agent(ctx, device: str, build_optimizer: bool,
                seed=None) -> "impala_agent.Agent":
    cfg = ctx.cfg
    return impala_agent.Agent(
        trajectory=cfg.trajectory, input_shape=cfg.model_input,
        num_action=cfg.model_output, lstm_hidden_size=cfg.lstm_size,
        discount_factor=cfg.discount_factor,
        start_learning_rate=cfg.start_learning_rate,
        end_learning_rate=cfg.end_learning_rate,
        learning_frame=cfg.learning_frame,
        baseline_loss_coef=cfg.baseline_loss_coef,
        entropy_coef=cfg.entropy_coef,
        gradient_clip_norm=cfg.gradient_clip_norm,
        reward_clipping=cfg.reward_clipping, device=device,
        build_optimizer=build_optimizer, seed=seed,
        model_arch=cfg.get("model_arch", "deep_conv"))


def learner(ctx: common.TrainerContext, supervisor=None) -> None:
    cfg, args = ctx.cfg, ctx.args
    queue = TrajectoryQueue(
        queue_schema_for("impala", cfg), cfg.num_actors, cfg.queue_size,
        role="learner", namespace=ctx.namespace, rank=ctx.rank,
        world_size=ctx.world_size)
    agent = build_agent(ctx, ctx.device, build_optimizer=True,
                        seed=args.seed)
    if args.restore:
        agent.load_weights(args.restore)
    from distributed_reinforcement_learning_amd.parallel.dist import broadcast_module
    broadcast_module(agent.model)  # rank-0 init (fixes reference C5 race)
    agent.setup_all_reduce()
    if ctx.rank == 0:
        agent.weight_publisher = WeightPublisher(ctx.weights_name,
                                                 agent.model.state_dict())
        agent.publish_weights()
    monitor = common.learner_supervision(ctx, queue, supervisor)
    writer = SummaryWriter(ctx.learner_logdir())
    timer = StageTimer()
    graphed = None
    pinned_views = None
    if torch.cuda.is_available() and not getattr(args, "no_graph", False):
        from distributed_reinforcement_learning_amd.runtime import (
            GraphedImpalaStep,
        )
        graphed = GraphedImpalaStep(agent, cfg.batch_size)
        # the queue fills the graphed step's pinned staging directly
        # (zero extra host copies); int64/bool fields go through a numpy
        # view-compatible dtype conversion inside the ring pop
        pinned_views = {k: v.numpy() for k, v in graphed.pinned.items()}
    train_step = 0
    log_every = max(1, args.publish_every) if graphed is None else 25
    # warm-up depth gate (reference train_impala.py:94-95: train only once
    # size > 3*batch_size). sample_batch blocks, so in steady state the
    # behavior is identical — the gate only delays the FIRST updates until
    # enough off-policy data has accumulated, matching the reference's
    # warm-up off-policy lag. GPU path only (the tiny CPU e2e runs would
    # wait out the deadline for depth their 2 actors never build), with a
    # deadline so a starved run degrades to blocking instead of hanging.
    if torch.cuda.is_available():
        gate_deadline = time.time() + 60.0
        while (queue.get_size() <= 3 * cfg.batch_size
               and time.time() < gate_deadline):
            if monitor is not None:
                supervisor.check()
            time.sleep(0.05)
    try:
        while args.max_steps <= 0 or train_step < args.max_steps:
            with timer.track("ingest"):
                if graphed is not None:
                    graphed.wait_pinned_free()
                batch = queue.sample_batch(cfg.batch_size, out=pinned_views)
            t0 = time.time()
            with timer.track("train"):
                if graphed is not None:
                    graphed.step()  # consumes the pinned staging
                    pi_loss = v_loss = entropy = lr = None
                else:
                    pi_loss, v_loss, entropy, lr = agent.train(
                        state=batch["state"], reward=batch["reward"],
                        action=batch["action"], done=batch["done"],
                        behavior_policy=batch["behavior_policy"],
                        previous_action=batch["previous_action"],
                        initial_h=batch["initial_h"],
                        initial_c=batch["initial_c"])
            train_step += 1
            if ctx.rank == 0 and train_step % args.publish_every == 0:
                with timer.track("publish"):
                    agent.publish_weights()
            ctx.maybe_checkpoint(agent)
            if monitor is not None and train_step % 50 == 0:
                supervisor.check()
            if ctx.rank == 0 and (graphed is None or train_step == 1
                                  or train_step % log_every == 0):
                if graphed is not None:
                    pi_loss, v_loss, entropy, lr = graphed.last_losses()
                step = agent.global_step
                writer.add_scalar("data/pi_loss", pi_loss, step)
                writer.add_scalar("data/value_loss", v_loss, step)
                writer.add_scalar("data/entropy", entropy, step)
                writer.add_scalar("data/lr", lr, step)
                writer.add_scalar("data/time", time.time() - t0, step)
                if train_step % 25 == 0:
                    frames_s = (cfg.batch_size * cfg.trajectory
                                * ctx.world_size
                                / max(timer.averages().get("train", 1e-9)
                                      + timer.averages().get("ingest", 0.0),
                                      1e-9))
                    writer.add_scalar("perf/env_frames_per_sec", frames_s,
                                      step)
                    print(f"[impala learner] step={step} "
                          f"frames/s={frames_s:,.0f} {timer.report()}",
                          flush=True)
    finally:
        writer.close()
        queue.close()
        if agent.weight_publisher:
            agent.weight_publisher.close()


def actor(ctx: common.TrainerContext, task: int) -> None:
    import torch
    torch.set_num_threads(1)  # batch-1 CPU inference; also avoids
    # the forked-child OpenMP deadlock (see trainers/common.py)
    cfg, args = ctx.cfg, ctx.args
    if int(cfg.get("envs_per_actor", 1)) > 1:
        return vector_actor(ctx, task)
    env_name = cfg.env[task]
    available_action = cfg.available_action[task]
    env = make_uint8_env(env_name, num_actions=cfg.model_output,
                         seed=(args.seed or 0) + task)
    queue = TrajectoryQueue(
        queue_schema_for("impala", cfg), cfg.num_actors, cfg.queue_size,
        role="actor", namespace=ctx.namespace, actor_task=task,
        world_size=ctx.world_size)
    agent = build_agent(ctx, "cpu", build_optimizer=False,
                        seed=(args.seed or 0) + 1000 + task)
    agent.weight_subscriber = WeightSubscriber(ctx.weights_name,
                                               agent.model.state_dict())
    agent.weight_subscriber.wait_for_first()
    writer = SummaryWriter(ctx.actor_logdir(task))
    traj = UnrolledTrajectory()

    state = env.reset()
    previous_action = 0
    h = np.zeros(cfg.lstm_size, dtype=np.float32)
    c = np.zeros(cfg.lstm_size, dtype=np.float32)
    episode, score, episode_step = 0, 0.0, 0
    prob_sum, prob_n = 0.0, 0
    unrolls = 0
    try:
        while args.max_unrolls <= 0 or unrolls < args.max_unrolls:
            traj.initialize()
            agent.parameter_sync()
            for _ in range(cfg.trajectory):
                action, policy, max_prob, nh, nc = \
                    agent.get_policy_and_action(state, previous_action, h, c)
                env_action = action % available_action
                next_state, reward, done, info = env.step(env_action)
                if info.get("life_lost"):
                    reward, done = -1.0, True
                score += reward
                episode_step += 1
                prob_sum += max_prob
                prob_n += 1
                traj.append(state=state, next_state=next_state,
                            previous_action=previous_action, action=action,
                            reward=reward, done=done, behavior_policy=policy,
                            initial_h=h, initial_c=c)
                state, previous_action, h, c = next_state, action, nh, nc
                if done:
                    writer.add_scalar("data/score", score, episode)
                    writer.add_scalar("data/episode_step", episode_step,
                                      episode)
                    writer.add_scalar("data/max_prob",
                                      prob_sum / max(prob_n, 1), episode)
                    episode += 1
                    score, episode_step, prob_sum, prob_n = 0.0, 0, 0.0, 0
                    state = env.reset()
                    previous_action = 0
                    h = np.zeros(cfg.lstm_size, dtype=np.float32)
                    c = np.zeros(cfg.lstm_size, dtype=np.float32)
            queue.append_to_queue(task, **traj.stacked())
            unrolls += 1
    finally:
        writer.close()
        queue.close()


def vector_actor(ctx: common.TrainerContext, task: int) -> None:
    """E envs per actor process, ONE batched inference per tick (VERDICT
    r1 item 8: the batch-1 python loop feeds ~0.1% of learner capacity).
    Same behavior per env as the scalar loop — unrolls of length T with
    stored per-step LSTM state, life-loss shaping, weight pull once per
    unroll period — but the model forward amortizes over E envs
    (config key ``envs_per_actor``; ``actor_threads`` sizes the torch
    intra-op pool for the batched forward — a 256-core EPYC host has
    ~12 cores per actor process to spare at the reference's 20 actors)."""
    import torch
    threads = int(ctx.cfg.get("actor_threads", 1))
    torch.set_num_threads(threads)
    try:
        import threadpoolctl
        ctx._tp_limits = threadpoolctl.threadpool_limits(threads)
    except Exception:
        pass
    cfg, args = ctx.cfg, ctx.args
    E = int(cfg.get("envs_per_actor", 1))
    env_name = cfg.env[task]
    available_action = cfg.available_action[task]
    envs = [make_uint8_env(env_name, num_actions=cfg.model_output,
                           seed=(args.seed or 0) + task * 1000 + e)
            for e in range(E)]
    queue = TrajectoryQueue(
        queue_schema_for("impala", cfg), cfg.num_actors, cfg.queue_size,
        role="actor", namespace=ctx.namespace, actor_task=task,
        world_size=ctx.world_size)
    agent = build_agent(ctx, "cpu", build_optimizer=False,
                        seed=(args.seed or 0) + 1000 + task)
    agent.weight_subscriber = WeightSubscriber(ctx.weights_name,
                                               agent.model.state_dict())
    agent.weight_subscriber.wait_for_first()
    writer = SummaryWriter(ctx.actor_logdir(task))

    H = cfg.lstm_size
    states = np.stack([env.reset() for env in envs])
    prev_action = np.zeros(E, dtype=np.int64)
    h = np.zeros((E, H), dtype=np.float32)
    c = np.zeros((E, H), dtype=np.float32)
    trajs = [UnrolledTrajectory() for _ in range(E)]
    for t in trajs:
        t.initialize()
    episode = 0
    score = np.zeros(E)
    episode_step = np.zeros(E, dtype=np.int64)
    unrolls = 0
    steps_in_unroll = 0
    try:
        while args.max_unrolls <= 0 or unrolls < args.max_unrolls * E:
            if steps_in_unroll == 0:
                agent.parameter_sync()
            actions, policies, max_probs, nh, nc = \
                agent.get_policy_and_action_batch(states, prev_action, h, c)
            for e, env in enumerate(envs):
                env_action = int(actions[e]) % available_action
                next_state, reward, done, info = env.step(env_action)
                if info.get("life_lost"):
                    reward, done = -1.0, True
                score[e] += reward
                episode_step[e] += 1
                trajs[e].append(
                    state=states[e], next_state=next_state,
                    previous_action=int(prev_action[e]),
                    action=int(actions[e]), reward=reward, done=done,
                    behavior_policy=policies[e], initial_h=h[e],
                    initial_c=c[e])
                states[e], prev_action[e] = next_state, actions[e]
                h[e], c[e] = nh[e], nc[e]
                if done:
                    writer.add_scalar("data/score", score[e], episode)
                    writer.add_scalar("data/episode_step",
                                      episode_step[e], episode)
                    episode += 1
                    score[e], episode_step[e] = 0.0, 0
                    states[e] = env.reset()
                    prev_action[e] = 0
                    h[e] = 0.0
                    c[e] = 0.0
            steps_in_unroll += 1
            if steps_in_unroll == cfg.trajectory:
                for e in range(E):
                    queue.append_to_queue(task, **trajs[e].stacked())
                    trajs[e].initialize()
                unrolls += E
                steps_in_unroll = 0
    finally:
        writer.close()
        queue.close()


def main(argv=None) -> None:
    common.run("impala", learner, actor, argv)
