"""R2D2 trainer loops (reference train_r2d2.py).

Learner (reference :87-165): ingest arrived sequences, score per-sequence
priority |mean TD| in one forward, PER sample of batch_size sequences,
IS-weighted recurrent update with burn-in, priority refresh, main->target
sync every 2500 steps (:164-165).

Actor (reference :167-251): POMDP env (20% blanked single frames), carries
h/c across steps, stores per-step state for the sequence, epsilon =
1/(0.01*episode+1) (:233); no action modulo (reference quirk, :200 — R2D2
acts directly in model action space).
"""

from __future__ import annotations

import time

import numpy as np

from distributed_reinforcement_learning_amd.agents import r2d2 as r2d2_agent
from distributed_reinforcement_learning_amd.envs import pomdp_uint8_env
from distributed_reinforcement_learning_amd.parallel.queue import (
    TrajectoryQueue, queue_schema_for,
)
from distributed_reinforcement_learning_amd.parallel.weights import (
    WeightPublisher, WeightSubscriber,
)
from distributed_reinforcement_learning_amd.replay import Memory
from distributed_reinforcement_learning_amd.trainers import common
from distributed_reinforcement_learning_amd.utils import StageTimer, SummaryWriter
from distributed_reinforcement_learning_amd.utils.trajectory import FieldTrajectory

TARGET_SYNC_EVERY = 2500     # reference train_r2d2.py:164
MEMORY_CAPACITY = 100_000    # reference train_r2d2.py:91-92


def build_agent(ctx, device: str, build_optimizer: bool, seed=None):
    cfg = ctx.cfg
    return r2d2_agent.Agent(
        seq_len=cfg.seq_len, burn_in=cfg.burn_in,
        input_shape=cfg.model_input, num_action=cfg.model_output,
        lstm_size=cfg.lstm_size, discount_factor=cfg.discount_factor,
        start_learning_rate=cfg.start_learning_rate,
        end_learning_rate=cfg.end_learning_rate,
        learning_frame=cfg.learning_frame,
        gradient_clip_norm=cfg.gradient_clip_norm,
        reward_clipping=cfg.reward_clipping, device=device,
        build_optimizer=build_optimizer, seed=seed)


def learner(ctx: common.TrainerContext, supervisor=None) -> None:
    cfg, args = ctx.cfg, ctx.args
    queue = TrajectoryQueue(
        queue_schema_for("r2d2", cfg), cfg.num_actors, cfg.queue_size,
        role="learner", namespace=ctx.namespace, rank=ctx.rank,
        world_size=ctx.world_size)
    agent = build_agent(ctx, ctx.device, True, args.seed)
    if args.restore:
        agent.load_weights(args.restore)
    from distributed_reinforcement_learning_amd.parallel.dist import broadcast_module
    broadcast_module(agent.model)
    broadcast_module(agent.target_model)
    agent.setup_all_reduce()
    agent.main_to_target()
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
        from distributed_reinforcement_learning_amd.replay.gpu_memory import (
            GpuMemory,
        )
        H, W, C = cfg.model_input
        L, Hs = cfg.seq_len, cfg.lstm_size
        memory = GpuMemory(MEMORY_CAPACITY, fields={
            "state": ((L, H, W, C), torch.uint8),
            "previous_action": ((L,), torch.int32),
            "action": ((L,), torch.int32),
            "reward": ((L,), torch.float32),
            "done": ((L,), torch.bool),
            "initial_h": ((L, Hs), torch.float32),
            "initial_c": ((L, Hs), torch.float32),
        }, device=ctx.device, seed=args.seed)
    else:
        memory = Memory(MEMORY_CAPACITY, seed=args.seed)
    train_step, buffer_steps = 0, 0
    graphed = None
    td_scorer = None
    min_warm = 2 * cfg.batch_size  # reference :122
    try:
        while args.max_steps <= 0 or train_step < args.max_steps:
            # one ingest per iteration (see trainers/apex.py note)
            need_data = buffer_steps < min_warm
            if queue.get_size() > 0 or need_data:
                with timer.track("ingest"):
                    u = queue.sample_batch(1)
                    if use_gpu_replay:
                        dev = {k: agent.to_device(v) for k, v in u.items()}
                        if td_scorer is None and not getattr(
                                args, "no_graph", False):
                            from distributed_reinforcement_learning_amd \
                                .runtime.replay_graphed import GraphedTdScore
                            td_scorer = GraphedTdScore(
                                agent, dev,
                                lambda i: agent.get_td_error_batch(
                                    i["state"], i["previous_action"],
                                    i["action"], i["initial_h"][:, 0],
                                    i["initial_c"][:, 0], i["reward"],
                                    i["done"], as_tensor=True))
                        if td_scorer is not None:
                            td = td_scorer.score(dev)
                            memory.add_batch(td, td_scorer.inputs)
                        else:
                            td = agent.get_td_error_batch(
                                dev["state"], dev["previous_action"],
                                dev["action"], dev["initial_h"][:, 0],
                                dev["initial_c"][:, 0], dev["reward"],
                                dev["done"], as_tensor=True)
                            memory.add_batch(td, dev)
                    else:
                        td = agent.get_td_error(
                            u["state"][0], u["previous_action"][0],
                            u["action"][0], u["initial_h"][0],
                            u["initial_c"][0], u["reward"][0], u["done"][0])
                        memory.add(td, {k: v[0] for k, v in u.items()})
                    buffer_steps += 1
            if buffer_steps < min_warm:
                continue
            t0 = time.time()
            if use_gpu_replay:
                if graphed is None and not getattr(args, "no_graph", False):
                    from distributed_reinforcement_learning_amd.runtime \
                        import GraphedReplayStep

                    def _loss_fn(b, w):
                        return agent.compute_sequence_loss(
                            b["state"], b["previous_action"], b["action"],
                            b["initial_h"][:, 0], b["initial_c"][:, 0],
                            b["reward"], b["done"], w)
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
                        loss, td_error = agent.train(
                            state=b["state"],
                            previous_action=b["previous_action"],
                            action=b["action"], h=b["initial_h"],
                            c=b["initial_c"], reward=b["reward"],
                            done=b["done"], weight=is_weight,
                            as_tensor=True)
                    with timer.track("per_update"):
                        memory.update_batch(idxs, td_error)
            else:
                with timer.track("sample"):
                    batch, idxs, is_weight = memory.sample(cfg.batch_size)
                    stacked = {k: np.stack([b[k] for b in batch])
                               for k in batch[0]}
                with timer.track("train"):
                    loss, td_error = agent.train(
                        state=stacked["state"],
                        previous_action=stacked["previous_action"],
                        action=stacked["action"], h=stacked["initial_h"],
                        c=stacked["initial_c"], reward=stacked["reward"],
                        done=stacked["done"], weight=is_weight)
                with timer.track("per_update"):
                    memory.update_batch(idxs, td_error)
            train_step += 1
            if train_step % TARGET_SYNC_EVERY == 0:
                agent.main_to_target()
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
                    print(f"[r2d2 learner] step={step} loss={loss:.5f} "
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
    env = pomdp_uint8_env(cfg.env[task], num_actions=cfg.model_output,
                          seed=(args.seed or 0) + task)
    queue = TrajectoryQueue(
        queue_schema_for("r2d2", cfg), cfg.num_actors, cfg.queue_size,
        role="actor", namespace=ctx.namespace, actor_task=task,
        world_size=ctx.world_size)
    agent = build_agent(ctx, "cpu", False, (args.seed or 0) + 1000 + task)
    agent.weight_subscriber = WeightSubscriber(ctx.weights_name,
                                               agent.model.state_dict())
    agent.weight_subscriber.wait_for_first()
    writer = SummaryWriter(ctx.actor_logdir(task))
    seq = FieldTrajectory(["state", "previous_action", "action", "reward",
                           "done", "initial_h", "initial_c"])

    state = env.reset()
    previous_action = 0
    h = np.zeros(cfg.lstm_size, dtype=np.float32)
    c = np.zeros(cfg.lstm_size, dtype=np.float32)
    episode, score, episode_step = 0, 0.0, 0
    q_sum = 0.0
    enqueued = 0
    try:
        while args.max_unrolls <= 0 or enqueued < args.max_unrolls:
            epsilon = 1.0 / (0.01 * episode + 1)  # reference :233
            action, q_a, nh, nc = agent.get_action(state, h, c,
                                                   previous_action, epsilon)
            next_state, reward, done, info = env.step(action)
            if info.get("life_lost"):
                reward, done = -1.0, True
            score += reward
            episode_step += 1
            q_sum += q_a
            seq.append(state=state, previous_action=previous_action,
                       action=action, reward=reward, done=done,
                       initial_h=h, initial_c=c)
            state, previous_action, h, c = next_state, action, nh, nc
            if len(seq) == cfg.seq_len:
                queue.append_to_queue(task, **seq.stacked())
                seq.initialize()
                enqueued += 1
                agent.parameter_sync()
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
                h = np.zeros(cfg.lstm_size, dtype=np.float32)
                c = np.zeros(cfg.lstm_size, dtype=np.float32)
    finally:
        writer.close()
        queue.close()


def vector_actor(ctx: common.TrainerContext, task: int) -> None:
    """E POMDP envs per actor process, ONE batched ε-greedy LSTM step per
    tick (``envs_per_actor``); per-env recurrent state carry, per-env
    sequence accumulators, same per-sequence enqueue + weight pull as the
    scalar loop (reference train_r2d2.py:167-251)."""
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
    envs = [pomdp_uint8_env(cfg.env[task], num_actions=cfg.model_output,
                            seed=(args.seed or 0) + task * 1000 + e)
            for e in range(E)]
    queue = TrajectoryQueue(
        queue_schema_for("r2d2", cfg), cfg.num_actors, cfg.queue_size,
        role="actor", namespace=ctx.namespace, actor_task=task,
        world_size=ctx.world_size)
    agent = build_agent(ctx, "cpu", False, (args.seed or 0) + 1000 + task)
    agent.weight_subscriber = WeightSubscriber(ctx.weights_name,
                                               agent.model.state_dict())
    agent.weight_subscriber.wait_for_first()
    writer = SummaryWriter(ctx.actor_logdir(task))
    H = cfg.lstm_size
    seqs = [FieldTrajectory(["state", "previous_action", "action",
                             "reward", "done", "initial_h", "initial_c"])
            for _ in range(E)]

    states = np.stack([env.reset() for env in envs])
    prev_action = np.zeros(E, dtype=np.int64)
    h = np.zeros((E, H), dtype=np.float32)
    c = np.zeros((E, H), dtype=np.float32)
    episode_n = np.zeros(E, dtype=np.int64)
    episode = 0
    score = np.zeros(E)
    episode_step = np.zeros(E, dtype=np.int64)
    q_sum = np.zeros(E)
    enqueued = 0
    try:
        while args.max_unrolls <= 0 or enqueued < args.max_unrolls * E:
            eps = 1.0 / (0.01 * episode_n + 1)  # reference :233, per env
            actions, q_a, nh, nc = agent.get_actions_batch(
                states, h, c, prev_action, eps)
            for e, env in enumerate(envs):
                next_state, reward, done, info = env.step(int(actions[e]))
                if info.get("life_lost"):
                    reward, done = -1.0, True
                score[e] += reward
                episode_step[e] += 1
                q_sum[e] += q_a[e]
                seqs[e].append(state=states[e],
                               previous_action=int(prev_action[e]),
                               action=int(actions[e]), reward=reward,
                               done=done, initial_h=h[e], initial_c=c[e])
                states[e], prev_action[e] = next_state, actions[e]
                h[e], c[e] = nh[e], nc[e]
                if len(seqs[e]) == cfg.seq_len:
                    queue.append_to_queue(task, **seqs[e].stacked())
                    seqs[e].initialize()
                    enqueued += 1
                    if enqueued % E == 0:
                        agent.parameter_sync()
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
                    h[e] = 0.0
                    c[e] = 0.0
    finally:
        writer.close()
        queue.close()


def main(argv=None) -> None:
    common.run("r2d2", learner, actor, argv)
