"""A3C feed-forward actor-critic (reference model/actor_critic.py).

conv stack + action embedding -> concat -> two 256 MLP heads
(policy softmax / scalar value). The reference builds twin evals for s and s'
with a shared scope (actor_critic.py:44-54); here that is just two forward
calls of the same module.

``VectorActorCritic`` is the 1-D-observation variant used by the CartPole
plumbing config (BASELINE config #1) where an image conv stack does not apply.
"""

from __future__ import annotations

import torch
import torch.nn as nn

from distributed_reinforcement_learning_amd.models.blocks import (
    ActionEmbedding, AtariConvStack, MLPHead,
)


class ActorCritic(nn.Module):
    def __init__(self, input_shape, num_action: int):
        super().__init__()
        h, w, c = input_shape
        assert (h, w) == (84, 84), "Atari conv stack expects 84x84 frames"
        self.num_action = num_action
        self.conv = AtariConvStack(c)
        self.action_emb = ActionEmbedding(num_action)
        feat = self.conv.out_features + self.action_emb.out_features
        self.policy_head = MLPHead(feat, [256, 256], num_action, "softmax")
        self.value_head = MLPHead(feat, [256, 256], 1, None)

    def forward(self, state: torch.Tensor, prev_action: torch.Tensor):
        """state: [N,84,84,C] normalized float (NHWC); prev_action: [N] int."""
        feat = torch.cat(
            [self.conv(state), self.action_emb(prev_action)], dim=1)
        policy = self.policy_head(feat)
        value = self.value_head(feat).squeeze(-1)
        return policy, value

    def logits_value(self, state: torch.Tensor, prev_action: torch.Tensor):
        """Pre-softmax logits + value — the fused A2C loss kernel's input
        (ops/a2c_op.py)."""
        feat = torch.cat(
            [self.conv(state), self.action_emb(prev_action)], dim=1)
        return self.policy_head.logits(feat), \
            self.value_head(feat).squeeze(-1)


class VectorActorCritic(nn.Module):
    """MLP actor-critic over 1-D observations (CartPole plumbing config)."""

    def __init__(self, input_shape, num_action: int, hidden: int = 256):
        super().__init__()
        (obs_dim,) = input_shape
        self.num_action = num_action
        self.obs_mlp = nn.Sequential(
            nn.Linear(obs_dim, hidden), nn.ReLU(),
            nn.Linear(hidden, hidden), nn.ReLU())
        self.action_emb = ActionEmbedding(num_action, hidden)
        self.policy_head = MLPHead(2 * hidden, [256, 256], num_action,
                                   "softmax")
        self.value_head = MLPHead(2 * hidden, [256, 256], 1, None)

    def forward(self, state: torch.Tensor, prev_action: torch.Tensor):
        state = state.to(self.obs_mlp[0].weight.dtype)
        feat = torch.cat(
            [self.obs_mlp(state), self.action_emb(prev_action)], dim=1)
        policy = self.policy_head(feat)
        value = self.value_head(feat).squeeze(-1)
        return policy, value

    def logits_value(self, state: torch.Tensor, prev_action: torch.Tensor):
        state = state.to(self.obs_mlp[0].weight.dtype)
        feat = torch.cat(
            [self.obs_mlp(state), self.action_emb(prev_action)], dim=1)
        return self.policy_head.logits(feat), \
            self.value_head(feat).squeeze(-1)
