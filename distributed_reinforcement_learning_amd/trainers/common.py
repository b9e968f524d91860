"""Shared trainer plumbing: CLI, topology wiring, spawn mode, run dirs.

CLI parity with the reference (train_impala.py:13-20): ``--job_name
{learner,actor}`` and ``--task N`` select the role; everything else comes from
config.json. Additions over the reference:

* ``--spawn``: launch the learner plus all actors from one command
  (the reference needs one shell command per process, README.md:25-69), with
  heartbeat-based actor respawn (parallel/heartbeat.py).
* torchrun-style multi-GPU learner: each rank owns its shard of actor rings;
  rank 0 publishes weights and logs.
* ``--max_steps`` bounds the run (tests/CI); ``--restore`` resumes from a
  checkpoint (capability the reference built but never called).
"""

from __future__ import annotations

import argparse
import os
import time
from typing import Callable, Optional

from distributed_reinforcement_learning_amd.utils import tunableop
tunableop.enable()  # before the first GEMM (learner hot path)

import torch

from distributed_reinforcement_learning_amd.config import (
    Config, default_config_path, load_config,
)
from distributed_reinforcement_learning_amd.parallel.dist import init_distributed
from distributed_reinforcement_learning_amd.parallel.heartbeat import (
    ActorSupervisor, HeartbeatMonitor,
)


def build_parser(algorithm: str) -> argparse.ArgumentParser:
    p = argparse.ArgumentParser(prog=f"train_{algorithm}")
    p.add_argument("--job_name", choices=["learner", "actor"],
                   default="learner")
    p.add_argument("--task", type=int, default=0)
    p.add_argument("--config", default=default_config_path())
    p.add_argument("--algorithm_block", default=algorithm,
                   help="config.json block name (e.g. a3c_cartpole)")
    p.add_argument("--spawn", action="store_true",
                   help="launch learner + all actors from this process")
    p.add_argument("--device", default=None,
                   help="learner device (default: cuda if available)")
    p.add_argument("--max_steps", type=int, default=0,
                   help="stop the learner after N train steps (0 = forever)")
    p.add_argument("--max_unrolls", type=int, default=0,
                   help="stop an actor after N unrolls (0 = forever)")
    p.add_argument("--runs_dir", default="runs")
    p.add_argument("--checkpoint_dir", default="checkpoints")
    p.add_argument("--checkpoint_every", type=int, default=500)
    p.add_argument("--publish_every", type=int, default=1,
                   help="publish weights every N train steps")
    p.add_argument("--restore", default=None)
    p.add_argument("--seed", type=int, default=None)
    p.add_argument("--no-graph", action="store_true",
                   help="disable hipGraph capture of the learner step")
    return p


class TrainerContext:
    def __init__(self, algorithm: str, args):
        self.algorithm = algorithm
        self.args = args
        self.cfg: Config = load_config(args.config, args.algorithm_block)
        self.local_rank = init_distributed()
        from distributed_reinforcement_learning_amd.parallel.dist import (
            rank, world_size,
        )
        self.rank = rank()
        self.world_size = world_size()
        self.namespace = f"{args.algorithm_block}_{self.cfg.server_port}"
        self.weights_name = f"drla_{self.namespace}_weights"
        if args.device:
            self.device = args.device
        elif torch.cuda.is_available():
            self.device = f"cuda:{self.local_rank}"
        else:
            self.device = "cpu"

    def learner_logdir(self) -> str:
        suffix = "" if self.rank == 0 else f"_rank{self.rank}"
        return os.path.join(self.args.runs_dir, "learner" + suffix)

    def actor_logdir(self, task: int) -> str:
        env_name = self.cfg.env[task]
        return os.path.join(self.args.runs_dir, env_name, f"actor_{task}")

    def checkpoint_path(self) -> str:
        os.makedirs(self.args.checkpoint_dir, exist_ok=True)
        return os.path.join(self.args.checkpoint_dir,
                            f"{self.args.algorithm_block}.pt")

    def maybe_checkpoint(self, agent) -> None:
        if (self.rank == 0 and self.args.checkpoint_every > 0
                and agent.global_step > 0
                and agent.global_step % self.args.checkpoint_every == 0):
            agent.save_weights(self.checkpoint_path())


def _actor_process_entry(algorithm: str, actor_fn: Callable, args,
                         task: int) -> None:
    """Module-level actor entry (picklable for the spawn start method).

    Runs outside the RCCL process group (scrub torchrun env) and
    single-threaded: batch-1 CPU inference wants one thread, and 20+ actors
    would oversubscribe the host otherwise.
    """
    for var in ("WORLD_SIZE", "RANK", "LOCAL_RANK"):
        os.environ.pop(var, None)
    torch.set_num_threads(1)
    # numpy's BLAS (the env pipeline's resize matmuls) defaults its pool
    # to nproc (256 on the EPYC hosts) PER PROCESS; with 20 actors inside
    # a 16-core cgroup quota that oversubscription collapses throughput
    # ~8x (measured r2). Clamp every pool to the actor's thread budget.
    try:
        import threadpoolctl
        global _TP_LIMITS  # keep alive: limits restore when GC'd
        _TP_LIMITS = threadpoolctl.threadpool_limits(1)
    except Exception:
        pass
    actor_args = argparse.Namespace(**vars(args))
    actor_args.spawn = False
    actor_args.job_name = "actor"
    actor_args.task = task
    actor_ctx = TrainerContext(algorithm, actor_args)
    actor_fn(actor_ctx, task)


def run(algorithm: str, learner_fn: Callable, actor_fn: Callable,
        argv=None) -> None:
    args = build_parser(algorithm).parse_args(argv)
    ctx = TrainerContext(algorithm, args)
    if args.spawn and ctx.rank == 0:
        import functools
        entry = functools.partial(_actor_process_entry, algorithm, actor_fn,
                                  args)
        sup = ActorSupervisor(entry, list(range(ctx.cfg.num_actors)))
        try:
            learner_fn(ctx, supervisor=sup)
        finally:
            sup.stop()
    elif args.job_name == "learner":
        learner_fn(ctx, supervisor=None)
    else:
        actor_fn(ctx, args.task)


def learner_supervision(ctx: TrainerContext, queue, supervisor
                        ) -> Optional[HeartbeatMonitor]:
    """Start actor processes (spawn mode) once the learner's rings exist."""
    if supervisor is None:
        return None
    monitor = HeartbeatMonitor(queue, timeout=120.0)
    supervisor.monitor = monitor
    supervisor.start()
    return monitor
