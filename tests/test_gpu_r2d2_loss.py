"""Fused R2D2 sequence-TD tail (K9, ops/hip/r2d2_loss.hip) vs the plain
torch fp32 composition of reference agent/r2d2.py:62-93 +
optimizer/burn_in.py:23-32. GPU-only."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _torch_reference(mq, tq, a, r, d, w, gamma, clip="abs_one"):
    """fp32 golden of the window tail (same math as agents/r2d2.py
    _sequence_losses after the burn-in slice)."""
    from distributed_reinforcement_learning_amd.agents.base import clip_rewards
    from distributed_reinforcement_learning_amd.algorithms import burn_in as rs
    from distributed_reinforcement_learning_amd.algorithms import dqn
    mq = mq.float()
    tq = tq.float()
    cr = clip_rewards(r, clip)
    disc = (~d).float() * gamma
    sav = dqn.take_state_action_value(mq[:, :-1], a[:, :-1])
    na = mq[:, 1:].argmax(dim=2)
    nsav = dqn.take_state_action_value(tq[:, 1:], na)
    tgt = rs.value_function_rescaling(
        (rs.inverse_value_function_rescaling(nsav) * disc[:, :-1]
         + cr[:, :-1]).detach())
    unweighted = ((tgt - sav) ** 2).mean(dim=1)
    loss = (unweighted * w).mean()
    td = (tgt - sav).mean(dim=1).abs()
    return loss, td


@pytest.mark.parametrize("W", [8, 15, 40])
def test_fused_r2d2_loss_matches_torch(W):
    from distributed_reinforcement_learning_amd.ops.r2d2_op import (
        fused_r2d2_loss,
    )
    torch.manual_seed(5)
    B, A = 6, 4
    mq = (torch.randn(B, W, A, device="cuda") * 2).to(
        torch.bfloat16).requires_grad_(True)
    tq = (torch.randn(B, W, A, device="cuda") * 2).to(torch.bfloat16)
    a = torch.randint(0, A, (B, W), device="cuda")
    r = torch.randn(B, W, device="cuda") * 2
    d = torch.rand(B, W, device="cuda") < 0.1
    w = torch.rand(B, device="cuda") + 0.5

    loss_f, td_f = fused_r2d2_loss(mq, tq, a, r, d, w, 0.997, "abs_one")
    loss_f.backward()

    mq2 = mq.detach().clone().requires_grad_(True)
    loss_t, td_t = _torch_reference(mq2, tq, a, r, d, w, 0.997)
    loss_t.backward()

    assert float(loss_f) == pytest.approx(float(loss_t), rel=2e-3)
    assert torch.allclose(td_f, td_t, atol=2e-3, rtol=2e-3)
    assert torch.allclose(mq.grad.float(), mq2.grad.float(), atol=2e-2,
                          rtol=2e-2), \
        (mq.grad.float() - mq2.grad.float()).abs().max()


def test_fused_r2d2_loss_soft_asymmetric_clip():
    from distributed_reinforcement_learning_amd.ops.r2d2_op import (
        fused_r2d2_loss,
    )
    torch.manual_seed(6)
    B, W, A = 4, 10, 6
    mq = torch.randn(B, W, A, device="cuda")
    tq = torch.randn(B, W, A, device="cuda")
    a = torch.randint(0, A, (B, W), device="cuda")
    r = torch.randn(B, W, device="cuda") * 4
    d = torch.zeros(B, W, dtype=torch.bool, device="cuda")
    w = torch.ones(B, device="cuda")
    loss_f, td_f = fused_r2d2_loss(mq, tq, a, r, d, w, 0.99,
                                   "soft_asymmetric")
    loss_t, td_t = _torch_reference(mq, tq, a, r, d, w, 0.99,
                                    "soft_asymmetric")
    assert float(loss_f) == pytest.approx(float(loss_t), rel=1e-4)
    assert torch.allclose(td_f, td_t, atol=1e-4, rtol=1e-4)


def test_agent_fused_path_matches_eager_composition():
    """Agent-level: compute_sequence_loss (fused K9 windows, no pad/cat)
    vs the eager _sequence_losses composition on the same GPU agent."""
    from distributed_reinforcement_learning_amd.agents import r2d2
    rng = np.random.default_rng(9)
    B, L, BI, A, H = 3, 12, 5, 4, 64
    agent = r2d2.Agent(seq_len=L, burn_in=BI, input_shape=[84, 84, 1],
                       num_action=A, lstm_size=H, discount_factor=0.997,
                       start_learning_rate=1e-4, end_learning_rate=0.0,
                       learning_frame=10 ** 9, gradient_clip_norm=40.0,
                       device="cuda:0", build_optimizer=False, seed=0)
    batch = dict(
        state=torch.as_tensor(rng.integers(
            0, 255, (B, L, 84, 84, 1), dtype=np.uint8)).cuda(),
        previous_action=torch.as_tensor(
            rng.integers(0, A, (B, L))).cuda(),
        action=torch.as_tensor(rng.integers(0, A, (B, L))).cuda(),
        reward=torch.as_tensor(
            rng.normal(size=(B, L)).astype(np.float32)).cuda(),
        done=torch.as_tensor(rng.random((B, L)) < 0.1).cuda(),
        h0=torch.zeros(B, H, device="cuda"),
        c0=torch.zeros(B, H, device="cuda"))
    w = torch.ones(B, device="cuda")

    assert agent._use_fused_tail()
    loss_f, td_f = agent.compute_sequence_loss(
        batch["state"], batch["previous_action"], batch["action"],
        batch["h0"], batch["c0"], batch["reward"], batch["done"], w)

    unweighted, tgt, sav = agent._sequence_losses(
        batch["state"], batch["previous_action"], batch["action"],
        batch["h0"], batch["c0"], batch["reward"], batch["done"],
        with_grad=False)
    loss_e = (unweighted * w).mean()
    td_e = (tgt - sav).mean(dim=1).abs()

    assert float(loss_f) == pytest.approx(float(loss_e), rel=0.05)
    assert torch.allclose(td_f, td_e, atol=0.05, rtol=0.05)


def test_dueling_head_kernel_matches_torch():
    """drla_dueling_head_fwd (one-launch no-grad head over the window)
    vs the torch composition on the same bf16 weights."""
    from distributed_reinforcement_learning_amd.models import R2D2LstmQ
    torch.manual_seed(13)
    m = R2D2LstmQ([84, 84, 1], 4, 64).cuda().bfloat16()
    from distributed_reinforcement_learning_amd import ops as _o
    ext = _o.require_ext()
    B, L, burn = 3, 12, 5
    h_all = torch.randn(B, L, 64, device="cuda")
    q_k = ext.dueling_head_fwd(
        h_all, m.trunk.weight.contiguous(), m.trunk.bias.contiguous(),
        m.out.weight.contiguous(), m.out.bias.contiguous(), burn)
    with torch.no_grad():
        q_t = m._head(h_all[:, burn:])
    assert q_k.shape == (B, L - burn, 4)
    assert torch.allclose(q_k.float(), q_t.float(), atol=3e-2, rtol=3e-2), \
        (q_k.float() - q_t.float()).abs().max()


def test_dueling_head_train_matches_torch():
    """Grad-carrying fused dueling head (drla_dhead_train_*) vs the torch
    composition: forward values and ALL five grads (dh, dWt, dbt, dWo,
    dbo)."""
    from distributed_reinforcement_learning_amd.ops.r2d2_op import (
        fused_dueling_head_train,
    )
    torch.manual_seed(17)
    N, IN, MID, A = 123, 64, 128, 4
    trunk = torch.nn.Linear(IN, MID).cuda().bfloat16()
    out = torch.nn.Linear(MID, A + 1).cuda().bfloat16()
    h = (torch.randn(N, IN, device="cuda")).requires_grad_(True)

    q = fused_dueling_head_train(h, trunk, out)
    g = torch.randn(N, A, device="cuda").to(torch.bfloat16)
    q.backward(g)

    # fp32 golden on the SAME bf16 weight values (the fused kernel
    # accumulates in f32 and rounds outputs to bf16 once; a bf16 torch
    # chain would add its own rounding at every step)
    h2 = h.detach().clone().requires_grad_(True)
    trunk2 = torch.nn.Linear(IN, MID).cuda()
    out2 = torch.nn.Linear(MID, A + 1).cuda()
    with torch.no_grad():
        trunk2.weight.copy_(trunk.weight.float())
        trunk2.bias.copy_(trunk.bias.float())
        out2.weight.copy_(out.weight.float())
        out2.bias.copy_(out.bias.float())
    x = torch.relu(trunk2(h2))
    y = out2(x)
    q2 = y[:, :A] - y[:, A:]
    q2.backward(g.float())

    assert torch.allclose(q.float(), q2, atol=4e-2, rtol=4e-2), \
        (q.float() - q2).abs().max()
    assert torch.allclose(h.grad, h2.grad, atol=6e-2, rtol=5e-2), \
        (h.grad - h2.grad).abs().max()
    pairs = [(trunk.weight.grad, trunk2.weight.grad, 0.5),
             (trunk.bias.grad, trunk2.bias.grad, 0.3),
             (out.weight.grad, out2.weight.grad, 0.5),
             (out.bias.grad, out2.bias.grad, 0.3)]
    for a, b, tol in pairs:
        assert a is not None and b is not None
        assert torch.allclose(a.float(), b.float(), atol=tol, rtol=0.05), \
            (a.float() - b.float()).abs().max()
