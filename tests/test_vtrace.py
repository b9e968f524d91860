"""V-trace numeric golden tests: torch implementation vs an independent numpy
re-derivation of the recursion in reference optimizer/vtrace.py:71-103."""

import numpy as np
import torch

from distributed_reinforcement_learning_amd.algorithms import vtrace


def numpy_vtrace(log_rhos, discounts, rewards, values, bootstrap,
                 clip_rho=1.0, clip_c=1.0):
    B, T = log_rhos.shape
    rhos = np.exp(log_rhos)
    crho = np.minimum(clip_rho, rhos)
    cs = np.minimum(clip_c, rhos)
    vtp1 = np.concatenate([values[:, 1:], bootstrap[:, None]], axis=1)
    deltas = crho * (rewards + discounts * vtp1 - values)
    vs_minus_v = np.zeros_like(values)
    acc = np.zeros(B)
    for t in reversed(range(T)):
        acc = deltas[:, t] + discounts[:, t] * cs[:, t] * acc
        vs_minus_v[:, t] = acc
    return vs_minus_v + values, crho


def _rand_case(B=5, T=18, A=7, seed=0):
    rng = np.random.default_rng(seed)
    logits_b = rng.normal(size=(B, T, A))
    logits_t = rng.normal(size=(B, T, A))
    softmax = lambda x: np.exp(x) / np.exp(x).sum(-1, keepdims=True)
    mu, pi = softmax(logits_b), softmax(logits_t)
    actions = rng.integers(0, A, size=(B, T))
    discounts = (rng.random((B, T)) > 0.1) * 0.99
    rewards = rng.normal(size=(B, T))
    values = rng.normal(size=(B, T))
    next_values = np.concatenate(
        [values[:, 1:], rng.normal(size=(B, 1))], axis=1)
    return mu, pi, actions, discounts, rewards, values, next_values


def test_from_softmax_matches_numpy_golden():
    mu, pi, actions, discounts, rewards, values, next_values = _rand_case()
    lp_pi = np.log(np.take_along_axis(pi, actions[..., None], 2)[..., 0])
    lp_mu = np.log(np.take_along_axis(mu, actions[..., None], 2)[..., 0])
    ref_vs, ref_rho = numpy_vtrace(lp_pi - lp_mu, discounts, rewards, values,
                                   next_values[:, -1])
    t = lambda x: torch.as_tensor(x, dtype=torch.float64)
    vs, rho = vtrace.from_softmax(
        t(mu), t(pi), torch.as_tensor(actions), t(discounts), t(rewards),
        t(values), t(next_values))
    np.testing.assert_allclose(vs.numpy(), ref_vs, rtol=1e-10, atol=1e-10)
    np.testing.assert_allclose(rho.numpy(), ref_rho, rtol=1e-10, atol=1e-10)


def test_vs_and_rho_carry_no_grad():
    mu, pi, actions, discounts, rewards, values, next_values = _rand_case()
    t32 = lambda x: torch.as_tensor(x, dtype=torch.float32)
    v = t32(values).requires_grad_(True)
    vs, rho = vtrace.from_softmax(
        t32(mu), t32(pi), torch.as_tensor(actions), t32(discounts),
        t32(rewards), v, t32(next_values))
    assert not vs.requires_grad and not rho.requires_grad


def test_on_policy_reduces_to_n_step_return():
    """With rho == c == 1 (on-policy), vs is the n-step Bellman target."""
    B, T = 3, 6
    rng = np.random.default_rng(1)
    rewards = rng.normal(size=(B, T))
    values = rng.normal(size=(B, T))
    bootstrap = rng.normal(size=B)
    gamma = 0.9
    discounts = np.full((B, T), gamma)
    vs, _ = numpy_vtrace(np.zeros((B, T)), discounts, rewards, values,
                         bootstrap)
    # direct n-step return: vs_t = sum_k gamma^k r_{t+k} + gamma^{T-t} V_boot
    for b in range(B):
        for t0 in range(T):
            ret = 0.0
            for k in range(t0, T):
                ret += gamma ** (k - t0) * rewards[b, k]
            ret += gamma ** (T - t0) * bootstrap[b]
            np.testing.assert_allclose(vs[b, t0], ret, rtol=1e-9)


def test_split_data_windows():
    x = torch.arange(10).reshape(1, 10)
    f, m, l = vtrace.split_data(x)
    assert f.tolist() == [[0, 1, 2, 3, 4, 5, 6, 7]]
    assert m.tolist() == [[1, 2, 3, 4, 5, 6, 7, 8]]
    assert l.tolist() == [[2, 3, 4, 5, 6, 7, 8, 9]]


def test_loss_reductions_match_reference_semantics():
    """Sum reductions + 1e-8 guard (reference vtrace.py:105-126)."""
    B, T, A = 2, 4, 3
    rng = np.random.default_rng(2)
    softmax = torch.softmax(torch.as_tensor(rng.normal(size=(B, T, A))), -1)
    actions = torch.as_tensor(rng.integers(0, A, (B, T)))
    adv = torch.as_tensor(rng.normal(size=(B, T)))
    pg = vtrace.compute_policy_gradient_loss(softmax, actions, adv)
    sel = softmax.gather(2, actions.unsqueeze(-1)).squeeze(-1)
    expected = -(torch.log(sel + 1e-8) * adv).sum()
    assert torch.allclose(pg, expected)

    vs = torch.as_tensor(rng.normal(size=(B, T)))
    val = torch.as_tensor(rng.normal(size=(B, T)))
    assert torch.allclose(vtrace.compute_baseline_loss(vs, val),
                          0.5 * (vs - val).pow(2).sum())

    ent = vtrace.compute_entropy_loss(softmax)
    expected_ent = -((-softmax * softmax.log()).sum(-1)).sum()
    assert torch.allclose(ent, expected_ent)


def test_vtrace_minimum_trajectory_length():
    """T=3 is the smallest trajectory the first/middle/last windowing
    admits (Tp = T-2 = 1); the whole CPU loss composition must hold."""
    import torch
    from distributed_reinforcement_learning_amd.agents import impala
    agent = impala.Agent(
        trajectory=3, input_shape=[84, 84, 4], num_action=4,
        lstm_hidden_size=8, discount_factor=0.99, start_learning_rate=1e-3,
        end_learning_rate=0.0, learning_frame=10 ** 9,
        baseline_loss_coef=1.0, entropy_coef=0.01, gradient_clip_norm=40.0,
        reward_clipping="abs_one", device="cpu", seed=0)
    rng = np.random.default_rng(0)
    B, T, A, H = 2, 3, 4, 8
    out = agent.train(
        state=rng.integers(0, 255, (B, T, 84, 84, 4), dtype=np.uint8),
        reward=rng.normal(size=(B, T)).astype(np.float32),
        action=rng.integers(0, A, (B, T)).astype(np.int32),
        done=np.zeros((B, T), dtype=bool),
        behavior_policy=np.full((B, T, A), 1 / A, dtype=np.float32),
        previous_action=rng.integers(0, A, (B, T)).astype(np.int32),
        initial_h=np.zeros((B, T, H), dtype=np.float32),
        initial_c=np.zeros((B, T, H), dtype=np.float32))
    assert all(np.isfinite(v) for v in out)
