"""Fused A2C loss pipeline (K6, ops/hip/a2c_loss.hip) vs the torch fp32
composition in algorithms/a2c.py (the golden). GPU-only."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _torch_reference(logits, value, next_value, a, r, d, gamma, c_bl,
                     c_ent, clip="abs_one"):
    from distributed_reinforcement_learning_amd.agents.base import clip_rewards
    from distributed_reinforcement_learning_amd.algorithms import a2c
    policy = torch.softmax(logits.float(), dim=-1)
    cr = clip_rewards(r, clip)
    disc = (~d).float() * gamma
    nv = next_value.float().detach()
    pi = a2c.compute_policy_loss(policy, a, value.float(), nv, disc, cr)
    bl = a2c.compute_baseline_loss(value.float(), nv, disc, cr)
    ent = a2c.compute_entropy_loss(policy)
    total = pi + bl * c_bl + ent * c_ent
    return pi, bl, ent, total


@pytest.mark.parametrize("A", [2, 18])
def test_fused_a2c_loss_matches_torch(A):
    from distributed_reinforcement_learning_amd.ops.a2c_op import (
        fused_a2c_loss,
    )
    torch.manual_seed(7)
    N = 32
    logits = (torch.randn(N, A, device="cuda") * 1.5).to(
        torch.bfloat16).requires_grad_(True)
    value = torch.randn(N, device="cuda").requires_grad_(True)
    next_value = torch.randn(N, device="cuda")
    a = torch.randint(0, A, (N,), device="cuda")
    r = torch.randn(N, device="cuda") * 2
    d = torch.rand(N, device="cuda") < 0.1

    pi_f, bl_f, ent_f, tot_f = fused_a2c_loss(
        logits, value, next_value, a, r, d, 0.997, "abs_one", 1.0, 0.05)
    tot_f.backward()

    logits2 = logits.detach().clone().requires_grad_(True)
    value2 = value.detach().clone().requires_grad_(True)
    pi_t, bl_t, ent_t, tot_t = _torch_reference(
        logits2, value2, next_value, a, r, d, 0.997, 1.0, 0.05)
    tot_t.backward()

    for f, t in [(pi_f, pi_t), (bl_f, bl_t), (ent_f, ent_t),
                 (tot_f, tot_t)]:
        assert float(f) == pytest.approx(float(t), rel=2e-3, abs=1e-4)
    assert torch.allclose(value.grad, value2.grad, atol=1e-4, rtol=1e-3)
    assert torch.allclose(logits.grad.float(), logits2.grad.float(),
                          atol=2e-2, rtol=2e-2), \
        (logits.grad.float() - logits2.grad.float()).abs().max()


def test_a3c_agent_fused_matches_cpu_golden():
    """Whole-agent: GPU compute_a2c_losses (K6) vs the same weights on the
    CPU torch path."""
    from distributed_reinforcement_learning_amd.agents import a3c
    rng = np.random.default_rng(1)
    N, A = 8, 4
    kw = dict(input_shape=[84, 84, 4], num_action=A, discount_factor=0.997,
              baseline_loss_coef=1.0, entropy_coef=0.05,
              start_learning_rate=1e-4, end_learning_rate=0.0,
              learning_frame=10 ** 9, gradient_clip_norm=40.0,
              reward_clipping="abs_one", build_optimizer=False, seed=5)
    g = a3c.Agent(device="cuda:0", **kw)
    c = a3c.Agent(device="cpu", **kw)
    c.model.load_state_dict(
        {k: v.float().cpu() for k, v in g.model.state_dict().items()})

    batch = dict(
        state=rng.integers(0, 255, (N, 84, 84, 4), dtype=np.uint8),
        next_state=rng.integers(0, 255, (N, 84, 84, 4), dtype=np.uint8),
        pa=rng.integers(0, A, N), a=rng.integers(0, A, N),
        r=rng.normal(size=N).astype(np.float32),
        d=rng.random(N) < 0.1)

    def run(agent):
        return agent.compute_a2c_losses(
            agent.frames_to_device(batch["state"]),
            agent.frames_to_device(batch["next_state"]),
            agent.to_device(batch["pa"], torch.int64),
            agent.to_device(batch["a"], torch.int64),
            agent.to_device(batch["r"], torch.float32),
            agent.to_device(batch["d"], torch.bool))

    out_g = [float(x) for x in run(g)]
    out_c = [float(x) for x in run(c)]
    # bf16 network forward vs fp32: loose tolerance, same math
    for a_, b_ in zip(out_g, out_c):
        assert a_ == pytest.approx(b_, rel=0.08, abs=0.05)
