"""Captured-RCCL-in-hipGraph proof harness.

Two modes, picked by WORLD_SIZE:

* world=2 (>=2 visible devices): ranks on cuda:0/cuda:1 — the FULL proof
  that the captured gather -> RCCL all-reduce -> fused-update optimizer
  graph works at world>1: rank-disjoint synthetic training must leave the
  flat parameter vectors bit-identical across ranks. The reference's whole
  topology is multi-process (/root/reference/train_impala.py:31-35).

* world=1 (1-GPU box): a REAL 1-rank RCCL communicator with
  DRLA_FORCE_DIST_GRAPH=1 — proves RCCL's collective launches capture and
  replay inside a hipGraph on this stack. This is the strongest proof a
  single MI355X allows: RCCL 2.26.6 hard-rejects two ranks on one device
  ("Duplicate GPU detected", ncclInvalidUsage — measured, r02) and the
  pool blocks CPX compute partitioning (rocm-smi --setcomputepartition cpx
  silently stays SPX), so multi-rank capture can only execute on a
  multi-GPU node. The world>1 delta (lockstep multi-rank replay) is
  covered by this same script in world=2 mode, which the gpu test suite
  runs automatically whenever >=2 devices are visible, and degrade is
  impossible: capture failure FAIL-FASTS (parallel/dist.py
  handle_capture_failure) instead of silently dropping to eager.

Launched by tests/test_gpu_dist.py as
    python -m torch.distributed.run --nnodes=1 --nproc-per-node {1,2} \
        --master-addr 127.0.0.1 --master-port P scripts/dist_graph_check.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import numpy as np
import torch


def main() -> None:
    from distributed_reinforcement_learning_amd.agents import impala
    from distributed_reinforcement_learning_amd.parallel import dist as pdist
    from distributed_reinforcement_learning_amd.parallel.dist import (
        broadcast_module,
    )
    from distributed_reinforcement_learning_amd.runtime import (
        GraphedImpalaStep,
    )

    ws = int(os.environ.get("WORLD_SIZE", "1"))
    if ws == 1:
        # init_distributed() no-ops at WORLD_SIZE<=1; build the 1-rank
        # RCCL communicator explicitly and force the distributed graph
        os.environ["DRLA_FORCE_DIST_GRAPH"] = "1"
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29539")
        os.environ.setdefault("RANK", "0")
        os.environ.setdefault("WORLD_SIZE", "1")
        torch.distributed.init_process_group(backend="nccl", rank=0,
                                             world_size=1)
        torch.cuda.set_device(0)
    else:
        pdist.init_distributed()
    world, rank = pdist.world_size(), pdist.rank()
    device = f"cuda:{torch.cuda.current_device()}"

    B, T, A, H = 8, 20, 18, 256
    agent = impala.Agent(
        trajectory=T, input_shape=[84, 84, 4], num_action=A,
        lstm_hidden_size=H, discount_factor=0.99, start_learning_rate=6e-4,
        end_learning_rate=0.0, learning_frame=10 ** 9,
        baseline_loss_coef=1.0, entropy_coef=0.05, gradient_clip_norm=40.0,
        reward_clipping="abs_one", device=device, seed=1234 + rank)
    broadcast_module(agent.model)
    agent.setup_all_reduce()
    if agent._all_reducer is None:
        # world=1 force mode: wire the reducer over the real 1-rank comm
        from distributed_reinforcement_learning_amd.parallel.dist import (
            FlatAllReducer,
        )
        agent._all_reducer = FlatAllReducer(agent.optimizer.flat_grads)

    graphed = GraphedImpalaStep(agent, B)
    assert graphed._distributed, "distributed graph path not taken"
    assert not graphed._eager_reduce, \
        "RCCL all-reduce did NOT capture (eager fallback engaged)"

    # rank-disjoint synthetic batches: only the all-reduce can keep the
    # parameter vectors identical at world>1
    rng = np.random.default_rng(500 + rank)
    batch = dict(
        state=rng.integers(0, 255, (B, T, 84, 84, 4), dtype=np.uint8),
        reward=rng.normal(size=(B, T)).astype(np.float32),
        action=rng.integers(0, A, (B, T)).astype(np.int32),
        done=(rng.random((B, T)) < 0.02),
        behavior_policy=np.full((B, T, A), 1.0 / A, dtype=np.float32),
        previous_action=rng.integers(0, A, (B, T)).astype(np.int32),
        initial_h=(rng.normal(size=(B, T, H)) * 0.1).astype(np.float32),
        initial_c=(rng.normal(size=(B, T, H)) * 0.1).astype(np.float32),
    )
    for _ in range(5):
        graphed.step(batch=batch)
    torch.cuda.synchronize()
    torch.distributed.barrier()

    flat = agent.optimizer.flat_params.detach().float()
    assert torch.isfinite(flat).all(), "non-finite parameters after replay"
    pi, bl, ent, _ = graphed.last_losses()
    assert np.isfinite([pi, bl, ent]).all(), "non-finite losses"
    if world > 1:
        gathered = [torch.empty_like(flat) for _ in range(world)]
        torch.distributed.all_gather(gathered, flat)
        if rank == 0:
            same = all(torch.equal(gathered[0], g) for g in gathered[1:])
            diff = max((gathered[0] - g).abs().max().item()
                       for g in gathered[1:])
            print(f"params_equal={same} max_diff={diff}", flush=True)
            assert same, f"rank parameter divergence: {diff}"
    if rank == 0:
        print(f"DIST_GRAPH_OK world={world}", flush=True)
    torch.distributed.barrier()


if __name__ == "__main__":
    main()
