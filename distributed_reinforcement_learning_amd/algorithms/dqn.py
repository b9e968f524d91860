"""n-step(=1) double-DQN target math (reference optimizer/dqn.py +
agent/apex.py:60-69)."""

from __future__ import annotations

from typing import Tuple

import torch


def take_state_action_value(state_value: torch.Tensor,
                            action: torch.Tensor) -> torch.Tensor:
    """Q[s, a] gather over the last dim (reference dqn.py:3-7's one-hot
    reduce)."""
    return state_value.gather(-1, action.long().unsqueeze(-1)).squeeze(-1)


@torch.no_grad()
def double_dqn_target(next_main_q: torch.Tensor, next_target_q: torch.Tensor,
                      rewards: torch.Tensor, discounts: torch.Tensor
                      ) -> Tuple[torch.Tensor, torch.Tensor]:
    """y = r + gamma * (1-d) * Q_target(s', argmax_a Q_main(s', a))
    (reference agent/apex.py:60-69). Returns (target, next_action)."""
    next_action = next_main_q.argmax(dim=-1)
    next_q = take_state_action_value(next_target_q, next_action)
    return rewards + discounts * next_q, next_action
