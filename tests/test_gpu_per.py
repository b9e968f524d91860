"""GPU PER segment tree vs the CPU SumTree/Memory golden. GPU-only."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _gpu_memory(cap=64, seed=0):
    from distributed_reinforcement_learning_amd.replay.gpu_memory import GpuMemory
    return GpuMemory(cap, fields={"x": ((3,), torch.float32)},
                     device="cuda:0", seed=seed)


def test_tree_total_matches_cpu():
    from distributed_reinforcement_learning_amd.replay import SumTree
    m = _gpu_memory(cap=32)
    cpu = SumTree(32)
    rng = np.random.default_rng(0)
    errs = rng.random(20).astype(np.float32) * 5
    m.add_batch(torch.as_tensor(errs, device="cuda"),
                {"x": torch.zeros(20, 3, device="cuda")})
    for e in errs:
        cpu.add(float((abs(e) + m.e) ** m.a), None)
    assert m.total() == pytest.approx(cpu.total(), rel=1e-5)


def test_ring_overwrite_and_update():
    m = _gpu_memory(cap=8)
    m.add_batch(torch.ones(12, device="cuda"),
                {"x": torch.arange(36, device="cuda",
                                   dtype=torch.float32).reshape(12, 3)})
    assert len(m) == 8
    # ring: rows hold samples 4..11
    assert float(m.data["x"][m.write % 8, 0]) == pytest.approx(4 * 3.0)
    # update one leaf to dominate
    idxs = torch.tensor([8 - 1 + 0], device="cuda")  # leaf 0
    m.update_batch(idxs, torch.full((1,), 1000.0, device="cuda"))
    expected_leaf = (1000.0 + m.e) ** m.a
    assert float(m.tree[8 - 1]) == pytest.approx(expected_leaf, rel=1e-4)


def test_sampling_tracks_priorities():
    m = _gpu_memory(cap=64, seed=1)
    errs = torch.full((32,), 0.01, device="cuda")
    errs[7] = 100.0
    m.add_batch(errs, {"x": torch.zeros(32, 3, device="cuda")})
    counts = torch.zeros(64)
    for _ in range(100):
        rows, idxs, w = m.sample(8)
        for r in rows.cpu():
            counts[r] += 1
        assert float(w.max()) == pytest.approx(1.0)
        assert (rows >= 0).all() and (rows < 64).all()
    assert counts[7] > counts.sum() * 0.5  # dominant priority wins


def test_gather_returns_stored_rows():
    m = _gpu_memory(cap=16)
    x = torch.arange(30, device="cuda", dtype=torch.float32).reshape(10, 3)
    m.add_batch(torch.ones(10, device="cuda"), {"x": x})
    rows = torch.tensor([2, 5], device="cuda")
    g = m.gather(rows)
    assert torch.equal(g["x"], x[[2, 5]])


def test_duplicate_updates_keep_tree_consistent():
    m = _gpu_memory(cap=16)
    m.add_batch(torch.ones(16, device="cuda"),
                {"x": torch.zeros(16, 3, device="cuda")})
    # duplicate index updates: root must equal sum of leaves afterwards
    idxs = torch.tensor([15, 15, 20, 20, 15], device="cuda")
    errs = torch.tensor([1.0, 2.0, 3.0, 4.0, 5.0], device="cuda")
    m.update_batch(idxs, errs)
    torch.cuda.synchronize()
    leaves = m.tree[15:31]
    assert float(m.tree[0]) == pytest.approx(float(leaves.sum()), rel=1e-4)


def test_fused_dqn_loss_matches_torch():
    from distributed_reinforcement_learning_amd.ops.dqn_op import fused_dqn_loss
    from distributed_reinforcement_learning_amd.algorithms import dqn
    torch.manual_seed(0)
    B, A = 32, 4
    mq = torch.randn(B, A, device="cuda", requires_grad=True)
    mq2 = mq.detach().clone().requires_grad_(True)
    nmq = torch.randn(B, A, device="cuda")
    ntq = torch.randn(B, A, device="cuda")
    a = torch.randint(0, A, (B,), device="cuda")
    r = torch.randn(B, device="cuda").clamp(-1, 1)
    disc = (torch.rand(B, device="cuda") > 0.2).float() * 0.99
    w = torch.rand(B, device="cuda") + 0.1

    loss_f, td_f = fused_dqn_loss(mq, nmq, ntq, a, r, disc, w)
    loss_f.backward()

    target, _ = dqn.double_dqn_target(nmq, ntq, r, disc)
    sav = dqn.take_state_action_value(mq2, a)
    loss_t = (((target.detach() - sav) ** 2) * w).mean()
    loss_t.backward()
    assert float(loss_f) == pytest.approx(float(loss_t), rel=1e-4)
    assert torch.allclose(td_f, target - sav.detach(), atol=1e-5)
    assert torch.allclose(mq.grad, mq2.grad, atol=1e-5)


def test_fused_dqn_loss_bf16_path():
    from distributed_reinforcement_learning_amd.ops.dqn_op import fused_dqn_loss
    B, A = 8, 6
    mq = torch.randn(B, A, device="cuda").bfloat16().requires_grad_(True)
    loss, td = fused_dqn_loss(
        mq, torch.randn(B, A, device="cuda"),
        torch.randn(B, A, device="cuda"),
        torch.randint(0, A, (B,), device="cuda"),
        torch.randn(B, device="cuda"), torch.full((B,), 0.99,
                                                  device="cuda"),
        torch.ones(B, device="cuda"))
    loss.backward()
    assert mq.grad.dtype == torch.bfloat16
    assert torch.isfinite(loss)


def test_graphed_replay_step_trains():
    """GraphedReplayStep (Ape-X flavor): replays must train (weights move),
    refresh priorities (tree root changes) and produce finite losses."""
    import numpy as np
    from distributed_reinforcement_learning_amd.agents import apex
    from distributed_reinforcement_learning_amd.replay.gpu_memory import (
        GpuMemory,
    )
    from distributed_reinforcement_learning_amd.runtime import (
        GraphedReplayStep,
    )
    agent = apex.Agent(
        input_shape=[84, 84, 4], num_action=6, discount_factor=0.99,
        gradient_clip_norm=40.0, reward_clipping="abs_one",
        start_learning_rate=1e-4, end_learning_rate=0.0,
        learning_frame=10 ** 9, device="cuda:0", build_optimizer=True,
        seed=5)
    mem = GpuMemory(4096, fields={
        "state": ((84, 84, 4), torch.uint8),
        "next_state": ((84, 84, 4), torch.uint8),
        "previous_action": ((), torch.int32),
        "action": ((), torch.int32),
        "reward": ((), torch.float32),
        "done": ((), torch.bool)}, device="cuda:0", seed=5)
    rng = np.random.default_rng(2)
    T = 64
    u = {
        "state": torch.as_tensor(rng.integers(0, 255, (T, 84, 84, 4),
                                              dtype=np.uint8)).cuda(),
        "next_state": torch.as_tensor(
            rng.integers(0, 255, (T, 84, 84, 4), dtype=np.uint8)).cuda(),
        "previous_action": torch.as_tensor(
            rng.integers(0, 6, T).astype(np.int32)).cuda(),
        "action": torch.as_tensor(
            rng.integers(0, 6, T).astype(np.int32)).cuda(),
        "reward": torch.as_tensor(
            rng.normal(size=T).astype(np.float32)).cuda(),
        "done": torch.as_tensor(rng.random(T) < 0.05).cuda()}
    td = agent.get_td_error(u["state"], u["next_state"],
                            u["previous_action"], u["action"], u["reward"],
                            u["done"], as_tensor=True)
    mem.add_batch(td, u)

    def loss_fn(b, w):
        return agent.compute_distributed_loss(
            agent.frames_to_device(b["state"]),
            agent.frames_to_device(b["next_state"]),
            b["previous_action"].long(), b["action"].long(),
            b["reward"], b["done"], w)

    p_before = agent.optimizer.flat_params.detach().clone()
    tree_before = float(mem.tree[0])
    g = GraphedReplayStep(agent, mem, 16, loss_fn)
    # construction must not perturb weights or priorities (snapshot/restore)
    assert torch.equal(agent.optimizer.flat_params, p_before)
    assert float(mem.tree[0]) == pytest.approx(tree_before, rel=1e-4)
    losses = [float(g.step()) for _ in range(5)]
    assert all(np.isfinite(losses))
    assert not torch.equal(agent.optimizer.flat_params, p_before)
    assert float(mem.tree[0]) != pytest.approx(tree_before, rel=1e-6)


def test_interior_rebuild_repairs_drift():
    """drla_per_rebuild_level: corrupt the interior sums, rebuild, and the
    tree must again be consistent bottom-up (the float32 atomicAdd delta
    chains drift over millions of updates — ADVICE r1)."""
    import torch
    m = _gpu_memory(cap=64, seed=1)
    rng = np.random.default_rng(2)
    m.add_batch(torch.as_tensor(rng.random(50).astype(np.float32) * 3,
                                device="cuda"),
                {"x": torch.zeros(50, 3, device="cuda")})
    leaves = m.tree[m.capacity - 1:].clone()
    # corrupt every interior node
    m.tree[:m.capacity - 1] += torch.rand(m.capacity - 1, device="cuda")
    m.rebuild()
    assert torch.equal(m.tree[m.capacity - 1:], leaves)
    # every interior node equals the sum of its children
    tree = m.tree.cpu()
    for i in range(m.capacity - 1):
        assert float(tree[i]) == pytest.approx(
            float(tree[2 * i + 1] + tree[2 * i + 2]), rel=1e-6, abs=1e-6)


def test_multi_gather_matches_index_select():
    """drla_multi_gather (one-kernel all-field batch gather) vs per-field
    index_select, incl. non-16-divisible row strides."""
    import torch
    from distributed_reinforcement_learning_amd.replay.gpu_memory import (
        GpuMemory,
    )
    rng = np.random.default_rng(3)
    m = GpuMemory(32, fields={
        "frames": ((5, 16), torch.uint8),     # 80 B rows (16-divisible)
        "vec": ((15,), torch.float32),        # 60 B rows (byte path)
        "flag": ((7,), torch.bool),           # 7 B rows (byte path)
        "idx": ((), torch.int32),             # 4 B rows
    }, device="cuda:0", seed=4)
    n = 20
    m.add_batch(
        torch.as_tensor(rng.random(n).astype(np.float32), device="cuda"),
        {"frames": torch.randint(0, 255, (n, 5, 16), dtype=torch.uint8,
                                 device="cuda"),
         "vec": torch.randn(n, 15, device="cuda"),
         "flag": torch.rand(n, 7, device="cuda") > 0.5,
         "idx": torch.arange(n, dtype=torch.int32, device="cuda")})
    rows = torch.as_tensor(rng.integers(0, n, 12), device="cuda")
    out = m.gather(rows)
    for k, buf in m.data.items():
        ref = buf.index_select(0, rows)
        assert torch.equal(out[k], ref), k
