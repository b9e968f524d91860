"""A2C (1-step-TD advantage actor-critic) losses.

Capability-parity with reference optimizer/a2c.py (mean reductions, 1-step TD
advantage, entropy term returned as *negative* entropy so the trainer adds it
with entropy_coef — a2c.py:3-26).

One deliberate fix over the reference: the reference's policy loss multiplies
the advantage by the selected *probability* (a2c.py:22-25) instead of its log
— a known defect. SURVEY.md §7 directs replicating capabilities, not defects,
so this uses log pi(a|s) (standard policy gradient).
"""

from __future__ import annotations

import torch


def compute_entropy_loss(policy: torch.Tensor) -> torch.Tensor:
    """-mean_t entropy_t over the batch ([N,A] softmax)."""
    # clamp inside the log: softmax underflow gives policy == 0 exactly
    # and 0 * log(0) is NaN; the correct limit of p*log p is 0
    entropy = (-policy * torch.log(policy.clamp_min(1e-30))).sum(dim=1)
    return -entropy.mean()


def compute_baseline_loss(value: torch.Tensor, next_value: torch.Tensor,
                          discounts: torch.Tensor,
                          reward: torch.Tensor) -> torch.Tensor:
    """mean (r + gamma * V(s') - V(s))^2 with V(s') detached."""
    diff = reward + discounts * next_value.detach() - value
    return (diff * diff).mean()


def compute_policy_loss(policy: torch.Tensor, action: torch.Tensor,
                        value: torch.Tensor, next_value: torch.Tensor,
                        discounts: torch.Tensor,
                        reward: torch.Tensor) -> torch.Tensor:
    """-mean adv * log pi(a|s), adv = r + gamma V(s') - V(s), detached."""
    sel = policy.gather(1, action.long().unsqueeze(-1)).squeeze(-1)
    advantage = (reward + discounts * next_value - value).detach()
    return -(advantage * torch.log(sel + 1e-8)).mean()
