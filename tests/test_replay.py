import numpy as np
import pytest

from distributed_reinforcement_learning_amd.replay import (
    LocalBuffer, Memory, SumTree,
)


class NaiveSumTree:
    """O(n) reference for the vectorized SumTree."""

    def __init__(self, capacity):
        self.capacity = capacity
        self.p = np.zeros(capacity)
        self.write = 0

    def add(self, p):
        self.p[self.write] = p
        self.write = (self.write + 1) % self.capacity

    def retrieve(self, s):
        cum = np.cumsum(self.p)
        # mirror tree descent: first leaf whose inclusive prefix >= s
        # (right-branch taken when s > left-subtree sum)
        for i, c in enumerate(cum):
            if s <= c:
                return i
        return len(self.p) - 1


def test_sum_tree_total_and_update():
    t = SumTree(8)
    idxs = [t.add(p, f"d{i}") for i, p in enumerate([1, 2, 3, 4])]
    assert t.total() == 10
    t.update(idxs[0], 5)
    assert t.total() == 14
    # batched update with duplicate index: later wins
    t.update_batch(np.array([idxs[1], idxs[1]]), np.array([7.0, 9.0]))
    assert t.total() == pytest.approx(5 + 9 + 3 + 4)


def test_sum_tree_retrieve_matches_naive():
    rng = np.random.default_rng(0)
    cap = 16
    t = SumTree(cap)
    naive = NaiveSumTree(cap)
    ps = rng.random(cap) + 0.01
    for i, p in enumerate(ps):
        t.add(p, i)
        naive.add(p)
    queries = rng.random(100) * t.total()
    got = t.retrieve_batch(queries) - (cap - 1)
    want = np.array([naive.retrieve(q) for q in queries])
    np.testing.assert_array_equal(got, want)


def test_sum_tree_ring_overwrite():
    t = SumTree(4)
    for i in range(6):
        t.add(1.0, i)
    # capacity 4: entries 2..5 remain
    datas = sorted(d for d in t.data)
    assert datas == [2, 3, 4, 5]
    assert t.n_entries == 4


def test_memory_priority_exponent_and_weights():
    m = Memory(64, seed=0)
    errors = np.array([0.0, 1.0, 10.0])
    for e in errors:
        m.add(e, f"s{e}")
    expected_p = (np.abs(errors) + Memory.e) ** Memory.a
    assert m.tree.total() == pytest.approx(expected_p.sum())
    batch, idxs, w = m.sample(3)
    assert len(batch) == 3 and len(idxs) == 3
    assert w.max() == pytest.approx(1.0)
    assert m.beta == pytest.approx(0.401)


def test_memory_sampling_tracks_priorities():
    """High-priority items must be sampled ~proportionally more."""
    m = Memory(128, seed=1)
    m.add(0.01, "low")
    m.add(100.0, "high")
    counts = {"low": 0, "high": 0}
    for _ in range(300):
        batch, _, _ = m.sample(2)
        for b in batch:
            counts[b] += 1
    assert counts["high"] > counts["low"] * 5


def test_memory_update_changes_distribution():
    m = Memory(8, seed=2)
    m.add(1.0, "a")
    m.add(1.0, "b")
    _, idxs, _ = m.sample(2)
    m.update(idxs[0], 1000.0)
    # (1000 + e)^0.6 ~ 63.1 plus the untouched leaf
    total = m.tree.total()
    assert total == pytest.approx((1000.0 + Memory.e) ** Memory.a
                                  + (1.0 + Memory.e) ** Memory.a)


def test_local_buffer_sample():
    lb = LocalBuffer(10, seed=0)
    for i in range(15):
        lb.append(i, i + 1, 0, 1, 0.5, False)
    assert len(lb) == 10
    s = lb.sample(4)
    assert len(s["state"]) == 4
    # holds only the last 10
    assert all(st >= 5 for st in s["state"])


def test_memory_state_dict_roundtrip():
    m = Memory(16, seed=3)
    for i in range(10):
        m.add(float(i), i)
    sd = m.state_dict()
    m2 = Memory(16)
    m2.load_state_dict(sd)
    assert m2.tree.total() == pytest.approx(m.tree.total())
    assert m2.tree.n_entries == m.tree.n_entries


def test_refresh_master_after_out_of_band_param_rewrite():
    """bf16 model copy + fp32 master: an out-of-band parameter rewrite
    (rank-0 broadcast, restore) must be followed by refresh_master(), or
    the next update runs on the stale master and silently reverts the
    rewrite (the world>1 divergence bug fixed in r2 — setup_all_reduce
    and load_weights now call it)."""
    import torch
    from distributed_reinforcement_learning_amd.ops.optim import FusedRMSProp

    p = torch.nn.Parameter(torch.randn(64).bfloat16())
    opt = FusedRMSProp([p], lr=0.0, clip_norm=None)  # lr=0: pure sync test
    assert opt.mixed
    # simulate a broadcast: overwrite the bf16 copy out-of-band
    with torch.no_grad():
        new = torch.randn(64).bfloat16()
        opt.flat_params[:64].copy_(new)
    opt.refresh_master()
    opt.flat_grads.zero_()
    opt.step(lr=0.0)  # lr 0: step must be a no-op on the params
    assert torch.equal(p.detach(), new), \
        "stale master reverted the out-of-band rewrite"
