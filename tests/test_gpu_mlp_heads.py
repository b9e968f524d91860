"""Fused policy+value MLP heads vs the torch composition. GPU-only."""

import pytest
import torch

pytestmark = pytest.mark.gpu


def _make_heads(A=18, seed=0):
    from distributed_reinforcement_learning_amd.models.blocks import MLPHead
    torch.manual_seed(seed)
    ph = MLPHead(256, [256, 256], A, None).cuda().bfloat16()
    vh = MLPHead(256, [256, 256], 1, None).cuda().bfloat16()
    return ph, vh


class _Holder:
    def __init__(self, ph, vh):
        self.policy_head, self.value_head = ph, vh


@pytest.mark.parametrize("N,A", [(640, 18), (37, 6)])
def test_fused_heads_forward_backward_parity(N, A):
    from distributed_reinforcement_learning_amd.ops.mlp_heads_op import (
        fused_mlp_heads, heads_fusable,
    )
    ph, vh = _make_heads(A)
    ph2, vh2 = _make_heads(A)
    ph2.load_state_dict(ph.state_dict())
    vh2.load_state_dict(vh.state_dict())

    h = torch.randn(N, 256, device="cuda", requires_grad=True)
    h2 = h.detach().clone().requires_grad_(True)
    assert heads_fusable(_Holder(ph, vh), h)

    logits_f, value_f = fused_mlp_heads(h, ph, vh)
    logits_t = ph2.logits(h2)
    value_t = vh2.logits(h2).squeeze(-1).float()

    rel = (logits_f.float() - logits_t.float()).abs().mean() / \
        logits_t.float().abs().mean().clamp(min=1e-3)
    assert rel < 0.05, f"logits rel err {rel.item():.4f}"
    assert torch.allclose(value_f, value_t, atol=0.05, rtol=0.05)

    dlog = torch.randn_like(logits_t.float())
    dval = torch.randn(N, device="cuda")
    (logits_f.float() * dlog).sum().backward(retain_graph=True)
    (value_f * dval).sum().backward()
    (logits_t.float() * dlog).sum().backward(retain_graph=True)
    (value_t * dval).sum().backward()

    assert torch.allclose(h.grad, h2.grad, atol=0.05, rtol=0.05), \
        f"dh max err {(h.grad - h2.grad).abs().max().item()}"
    pairs = []
    for m1, m2 in [(ph, ph2), (vh, vh2)]:
        for l1, l2 in zip(list(m1.hidden) + [m1.out],
                          list(m2.hidden) + [m2.out]):
            pairs.append((l1.weight.grad, l2.weight.grad, "w"))
            pairs.append((l1.bias.grad, l2.bias.grad, "b"))
    for g1, g2, kind in pairs:
        assert g1 is not None and g2 is not None
        scale = g2.float().abs().mean().clamp(min=1e-4)
        rel = (g1.float() - g2.float()).abs().mean() / scale
        assert rel < 0.08, f"{kind} grad rel err {rel.item():.4f}"


def test_fused_heads_in_agent_train():
    """IMPALA flagship-shape agent exercises the fused-heads path."""
    import numpy as np
    from distributed_reinforcement_learning_amd.agents import impala
    agent = impala.Agent(
        trajectory=8, input_shape=[84, 84, 4], num_action=18,
        lstm_hidden_size=256, discount_factor=0.99,
        start_learning_rate=1e-3, end_learning_rate=0.0,
        learning_frame=10 ** 9, baseline_loss_coef=1.0, entropy_coef=0.05,
        gradient_clip_norm=40.0, reward_clipping="abs_one",
        device="cuda:0", seed=0)
    rng = np.random.default_rng(0)
    B, T, A, H = 4, 8, 18, 256
    for _ in range(2):
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
