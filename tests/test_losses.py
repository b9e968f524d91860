import numpy as np
import torch

from distributed_reinforcement_learning_amd.algorithms import a2c, burn_in, dqn


def test_rescaling_inverse_roundtrip():
    x = torch.linspace(-50, 50, 1001, dtype=torch.float64)
    h = burn_in.value_function_rescaling(x)
    back = burn_in.inverse_value_function_rescaling(h)
    assert torch.allclose(back, x, atol=1e-6)


def test_rescaling_known_values():
    x = torch.tensor([0.0, 3.0, -3.0], dtype=torch.float64)
    h = burn_in.value_function_rescaling(x, eps=1e-3)
    exp = torch.sign(x) * (torch.sqrt(x.abs() + 1) - 1) + 1e-3 * x
    assert torch.allclose(h, exp)


def test_double_dqn_target():
    torch.manual_seed(0)
    B, A = 8, 5
    next_main = torch.randn(B, A)
    next_target = torch.randn(B, A)
    r = torch.randn(B)
    d = torch.randint(0, 2, (B,)).bool()
    disc = (~d).float() * 0.99
    y, na = dqn.double_dqn_target(next_main, next_target, r, disc)
    for b in range(B):
        a_star = next_main[b].argmax()
        assert na[b] == a_star
        expected = r[b] + disc[b] * next_target[b, a_star]
        assert torch.allclose(y[b], expected)


def test_take_state_action_value():
    q = torch.arange(12.0).reshape(3, 4)
    a = torch.tensor([0, 3, 2])
    v = dqn.take_state_action_value(q, a)
    assert v.tolist() == [0.0, 7.0, 10.0]


def test_a2c_losses_mean_semantics():
    torch.manual_seed(1)
    N, A = 16, 3
    policy = torch.softmax(torch.randn(N, A), -1)
    action = torch.randint(0, A, (N,))
    value = torch.randn(N, requires_grad=True)
    next_value = torch.randn(N)
    r = torch.randn(N)
    disc = torch.full((N,), 0.99)

    bl = a2c.compute_baseline_loss(value, next_value, disc, r)
    diff = r + disc * next_value - value
    assert torch.allclose(bl, (diff ** 2).mean())

    pl = a2c.compute_policy_loss(policy, action, value, next_value, disc, r)
    sel = policy.gather(1, action.unsqueeze(1)).squeeze(1)
    adv = (r + disc * next_value - value).detach()
    assert torch.allclose(pl, -(adv * torch.log(sel + 1e-8)).mean())

    el = a2c.compute_entropy_loss(policy)
    ent = (-policy * policy.log()).sum(1)
    assert torch.allclose(el, -ent.mean())


def test_a2c_baseline_loss_does_not_backprop_next_value():
    value = torch.randn(4, requires_grad=True)
    next_value = torch.randn(4, requires_grad=True)
    loss = a2c.compute_baseline_loss(value, next_value,
                                     torch.full((4,), 0.9), torch.ones(4))
    loss.backward()
    assert next_value.grad is None
    assert value.grad is not None
