"""Ape-X trainer loops (reference train_apex.py).

Learner (reference :82-155), two phases per iteration:
  A. ingest — drain arrived unrolls, score TD errors in ONE batched forward
     (the reference scores per-transition on a batch of 1 unroll,
     :94-122), push into the prioritized Memory;
  B. train — stratified PER sample, IS-weighted double-DQN update, priority
     refresh, target sync every 100 steps (:151-152).

Actor (reference :157-231): epsilon = 1/(0.05*episode+1), LocalBuffer of 1e4
transitions, and once warm it enqueues a fresh RANDOM sample of `trajectory`
transitions every step (reference :207-217 — the queue carries uniform
re-samples, not contiguous unrolls; kept, but throttled to one enqueue per
env step as the reference does).
"""

from __future__ import annotations

import time

import numpy as np

from distributed_reinforcement_learning_amd.agents import apex as apex_agent
from distributed_reinforcement_learning_amd.envs import make_uint8_env
from distributed_reinforcement_learning_amd.parallel.queue import (
    TrajectoryQueue, queue_schema_for,
)
from distributed_reinforcement_learning_amd.parallel.weights import (
    WeightPublisher, WeightSubscriber,
)
from distributed_reinforcement_learning_amd.replay import LocalBuffer, Memory
from distributed_reinforcement_learning_amd.trainers import common
from distributed_reinforcement_learning_amd.utils import StageTimer, SummaryWriter

TARGET_SYNC_EVERY = 100       # reference train_apex.py:151
MEMORY_CAPACITY = 100_000     # reference train_apex.py:86
LOCAL_BUFFER_CAPACITY = 10_000  # reference train_apex.py:159-160
TRAIN_AFTER = 10              # reference train_apex.py:124


def build_agent(ctx, device: str, build_optimizer: bool, seed=None):
    cfg = ctx.cfg
    return apex_agent.Agent(
        input_shape=cfg.model_input, num_action=cfg.model_output,
        discount_factor=cfg.discount_factor,
        gradient_clip_norm=cfg.gradient_clip_norm,
        reward_clipping=cfg.reward_clipping,
        start_learning_rate=cfg.start_learning_rate,
        end_learning_rate=cfg.end_learning_rate,
        learning_frame=cfg.learning_frame, device=device,
        build_optimizer=build_optimizer, seed=seed)


def learner(ctx: common.TrainerContext, supervisor=None) -> None:
    cfg, args = ctx.cfg, ctx.args
    queue = TrajectoryQueue(
        queue_schema_for("apex", cfg), cfg.num_actors, cfg.queue_size,
        role="learner", namespace=ctx.namespace, rank=ctx.rank,
        world_size=ctx.world_size)
    agent = build_agent(ctx, ctx.device, True, args.seed)
    if args.restore:
        agent.load_weights(args.restore)
    from distributed_reinforcement_learning_amd.parallel.dist import broadcast_module
    broadcast_module(agent.model)
    broadcast_module(agent.target_model)
    agent.setup_all_reduce()
    agent.target_to_main()
    if ctx.rank == 0:
        agent.weight_publisher = WeightPublisher(ctx.weights_name,
                                                 agent.model.state_dict())
        agent.publish_weights()
    monitor = common.learner_supervision(ctx, queue, supervisor)
    writer = SummaryWriter(ctx.learner_logdir())
    timer = StageTimer()
    import torch
    use_gpu_replay = torch.cuda.is_available()
    if use_gpu_replay:
        # GPU-resident PER shard (replay/gpu_memory.py): segment tree,
        # priorities and payloads all on-device
        from distributed_reinforcement_learning_amd.replay.gpu_memory import (
            GpuMemory,
        )
        H, W, C = cfg.model_input
        memory = GpuMemory(MEMORY_CAPACITY, fields={
            "state": ((H, W, C), torch.uint8),
            "next_state": ((H, W, C), torch.uint8),
            "previous_action": ((), torch.int32),
            "action": ((), torch.int32),
            "reward": ((), torch.float32),
            "done": ((), torch.bool),
        }, device=ctx.device, seed=args.seed)
    else:
        memory = Memory(MEMORY_CAPACITY, seed=args.seed)
    train_step, buffer_steps = 0, 0
    graphed = None
    td_scorer = None
    try:
        while args.max_steps <= 0 or train_step < args.max_steps:
            # Phase A: ingest one arrived unroll per iteration (blocking
            # while the memory is cold) — the reference's 1:1 interleave
            # (train_apex.py:94-122); never drain-to-empty, or a fast actor
            # pool starves the train phase.
            need_data = buffer_steps <= TRAIN_AFTER
            if queue.get_size() > 0 or need_data:
                with timer.track("ingest"):
                    u = queue.sample_batch(1)
                    T = u["state"].shape[1]
                    if use_gpu_replay:
                        dev = {k: agent.to_device(v[0]) for k, v in u.items()}
                        if td_scorer is None and not getattr(
                                args, "no_graph", False):
                            from distributed_reinforcement_learning_amd \
                                .runtime.replay_graphed import GraphedTdScore
                            td_scorer = GraphedTdScore(
                                agent, dev,
                                lambda i: agent.get_td_error(
                                    i["state"], i["next_state"],
                                    i["previous_action"], i["action"],
                                    i["reward"], i["done"], as_tensor=True))
                        if td_scorer is not None:
                            td = td_scorer.score(dev)
                            memory.add_batch(td, td_scorer.inputs)
                        else:
                            td = agent.get_td_error(
                                dev["state"], dev["next_state"],
                                dev["previous_action"], dev["action"],
                                dev["reward"], dev["done"], as_tensor=True)
                            memory.add_batch(td, dev)
                    else:
                        td = agent.get_td_error(
                            u["state"][0], u["next_state"][0],
                            u["previous_action"][0], u["action"][0],
                            u["reward"][0], u["done"][0])
                        samples = [
                            (u["state"][0, t], u["next_state"][0, t],
                             u["previous_action"][0, t], u["action"][0, t],
                             u["reward"][0, t], u["done"][0, t])
                            for t in range(T)
                        ]
                        memory.add_batch(td, samples)
                    buffer_steps += 1
            if buffer_steps <= TRAIN_AFTER:
                continue
            # Phase B: PER train
            t0 = time.time()
            if use_gpu_replay:
                if graphed is None and not getattr(args, "no_graph", False):
                    # hipGraph-captured sample+train+priority-update
                    # (runtime/replay_graphed.py); built once the replay
                    # shard is warm
                    from distributed_reinforcement_learning_amd.runtime \
                        import GraphedReplayStep

                    def _loss_fn(b, w):
                        return agent.compute_distributed_loss(
                            agent.frames_to_device(b["state"]),
                            agent.frames_to_device(b["next_state"]),
                            b["previous_action"].long(),
                            b["action"].long(), b["reward"], b["done"], w)
                    graphed = GraphedReplayStep(agent, memory,
                                                cfg.batch_size, _loss_fn)
                if graphed is not None:
                    with timer.track("train"):
                        loss = graphed.step()
                else:
                    with timer.track("sample"):
                        rows, idxs, is_weight = memory.sample(
                            cfg.batch_size)
                        b = memory.gather(rows)
                    with timer.track("train"):
                        loss, td_error = agent.distributed_train(
                            b["state"], b["next_state"],
                            b["previous_action"], b["action"], b["reward"],
                            b["done"], is_weight, as_tensor=True)
                    with timer.track("per_update"):
                        memory.update_batch(idxs, td_error)
            else:
                with timer.track("sample"):
                    batch, idxs, is_weight = memory.sample(cfg.batch_size)
                    stacked = [np.stack([b[i] for b in batch])
                               for i in range(6)]
                with timer.track("train"):
                    loss, td_error = agent.distributed_train(
                        *stacked, is_weight)
                with timer.track("per_update"):
                    memory.update_batch(idxs, td_error)
            train_step += 1
            if train_step % TARGET_SYNC_EVERY == 0:
                agent.target_to_main()
            if ctx.rank == 0 and train_step % args.publish_every == 0:
                agent.publish_weights()
            ctx.maybe_checkpoint(agent)
            if monitor is not None and train_step % 50 == 0:
                supervisor.check()
            # loss may be a device tensor (GPU path): convert only on the
            # logging cadence — a per-step .item() blocks the pipeline for
            # ~11 ms of wake latency
            if ctx.rank == 0 and (train_step % 25 == 0 or train_step == 1):
                step = agent.global_step
                writer.add_scalar("data/loss", float(loss), step)
                writer.add_scalar("data/time", time.time() - t0, step)
                if train_step % 50 == 0:
                    print(f"[apex learner] step={step} loss={float(loss):.4f} "
                          f"{timer.report()}", flush=True)
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
    env = make_uint8_env(cfg.env[task], num_actions=cfg.model_output,
                         seed=(args.seed or 0) + task)
    available_action = cfg.available_action[task]
    queue = TrajectoryQueue(
        queue_schema_for("apex", cfg), cfg.num_actors, cfg.queue_size,
        role="actor", namespace=ctx.namespace, actor_task=task,
        world_size=ctx.world_size)
    agent = build_agent(ctx, "cpu", False, (args.seed or 0) + 1000 + task)
    agent.weight_subscriber = WeightSubscriber(ctx.weights_name,
                                               agent.model.state_dict())
    agent.weight_subscriber.wait_for_first()
    writer = SummaryWriter(ctx.actor_logdir(task))
    local = LocalBuffer(LOCAL_BUFFER_CAPACITY, seed=(args.seed or 0) + task)
    rng = np.random.default_rng((args.seed or 0) + task)

    state = env.reset()
    previous_action = 0
    episode, score, episode_step = 0, 0.0, 0
    q_sum = 0.0
    enqueued = 0
    T = cfg.trajectory
    try:
        while args.max_unrolls <= 0 or enqueued < args.max_unrolls:
            epsilon = 1.0 / (0.05 * episode + 1)  # reference :229
            action, q_values, q_a = agent.get_policy_and_action(
                state, previous_action, epsilon)
            env_action = action % available_action
            next_state, reward, done, info = env.step(env_action)
            if info.get("life_lost"):
                reward, done = -1.0, True
            score += reward
            episode_step += 1
            q_sum += q_a
            local.append(state, next_state, previous_action, action,
                         reward, done)
            state, previous_action = next_state, action
            if len(local) > 3 * T:
                s = local.sample(T)
                queue.append_to_queue(
                    task,
                    state=np.stack(s["state"]),
                    next_state=np.stack(s["next_state"]),
                    previous_action=np.asarray(s["previous_action"],
                                               np.int32),
                    action=np.asarray(s["action"], np.int32),
                    reward=np.asarray(s["reward"], np.float32),
                    done=np.asarray(s["done"], np.bool_))
                enqueued += 1
            if done:
                writer.add_scalar("data/score", score, episode)
                writer.add_scalar("data/episode_step", episode_step, episode)
                writer.add_scalar("data/epsilon", epsilon, episode)
                writer.add_scalar("data/avg_q",
                                  q_sum / max(episode_step, 1), episode)
                episode += 1
                score, episode_step, q_sum = 0.0, 0, 0.0
                state = env.reset()
                previous_action = 0
                agent.parameter_sync()  # reference pulls per episode (:177)
    finally:
        writer.close()
        queue.close()


def vector_actor(ctx: common.TrainerContext, task: int) -> None:
    """E envs per actor process with ONE batched ε-greedy forward per tick
    (``envs_per_actor`` — same redesign as trainers/impala.vector_actor);
    per-env LocalBuffer/episode state, same enqueue-a-random-sample
    behavior as the scalar loop (reference train_apex.py:157-231)."""
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
    available_action = cfg.available_action[task]
    envs = [make_uint8_env(cfg.env[task], num_actions=cfg.model_output,
                           seed=(args.seed or 0) + task * 1000 + e)
            for e in range(E)]
    queue = TrajectoryQueue(
        queue_schema_for("apex", cfg), cfg.num_actors, cfg.queue_size,
        role="actor", namespace=ctx.namespace, actor_task=task,
        world_size=ctx.world_size)
    agent = build_agent(ctx, "cpu", False, (args.seed or 0) + 1000 + task)
    agent.weight_subscriber = WeightSubscriber(ctx.weights_name,
                                               agent.model.state_dict())
    agent.weight_subscriber.wait_for_first()
    writer = SummaryWriter(ctx.actor_logdir(task))
    locals_ = [LocalBuffer(LOCAL_BUFFER_CAPACITY,
                           seed=(args.seed or 0) + task * 1000 + e)
               for e in range(E)]

    states = np.stack([env.reset() for env in envs])
    prev_action = np.zeros(E, dtype=np.int64)
    episode_n = np.zeros(E, dtype=np.int64)
    episode = 0
    score = np.zeros(E)
    episode_step = np.zeros(E, dtype=np.int64)
    q_sum = np.zeros(E)
    enqueued = 0
    T = cfg.trajectory
    sync_every = 0
    try:
        while args.max_unrolls <= 0 or enqueued < args.max_unrolls * E:
            eps = 1.0 / (0.05 * episode_n + 1)  # reference :229, per env
            actions, _, q_a = agent.get_actions_batch(states, prev_action,
                                                      eps)
            for e, env in enumerate(envs):
                next_state, reward, done, info = env.step(
                    int(actions[e]) % available_action)
                if info.get("life_lost"):
                    reward, done = -1.0, True
                score[e] += reward
                episode_step[e] += 1
                q_sum[e] += q_a[e]
                locals_[e].append(states[e], next_state,
                                  int(prev_action[e]), int(actions[e]),
                                  reward, done)
                states[e], prev_action[e] = next_state, actions[e]
                if len(locals_[e]) > 3 * T:
                    s = locals_[e].sample(T)
                    queue.append_to_queue(
                        task,
                        state=np.stack(s["state"]),
                        next_state=np.stack(s["next_state"]),
                        previous_action=np.asarray(s["previous_action"],
                                                   np.int32),
                        action=np.asarray(s["action"], np.int32),
                        reward=np.asarray(s["reward"], np.float32),
                        done=np.asarray(s["done"], np.bool_))
                    enqueued += 1
                if done:
                    writer.add_scalar("data/score", score[e], episode)
                    writer.add_scalar("data/episode_step",
                                      episode_step[e], episode)
                    writer.add_scalar("data/epsilon", eps[e], episode)
                    writer.add_scalar("data/avg_q",
                                      q_sum[e] / max(episode_step[e], 1),
                                      episode)
                    episode += 1
                    episode_n[e] += 1
                    score[e], episode_step[e], q_sum[e] = 0.0, 0, 0.0
                    states[e] = env.reset()
                    prev_action[e] = 0
            sync_every += 1
            if sync_every >= T:  # amortized weight pull (per-episode in
                sync_every = 0   # the scalar loop)
                agent.parameter_sync()
    finally:
        writer.close()
        queue.close()


def main(argv=None) -> None:
    common.run("apex", learner, actor, argv)
