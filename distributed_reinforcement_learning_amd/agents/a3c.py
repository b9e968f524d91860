"""A3C (distributed A2C) agent — capability-parity with reference
agent/a3c.py: twin eval of (s, s') through one shared network, 1-step-TD
advantage losses (optimizer/a2c.py), Adam + global-norm clip + polynomial LR.

Model is chosen by observation rank: conv actor-critic for image inputs
(reference model/actor_critic.py), MLP for vector inputs (CartPole plumbing
config, BASELINE config #1).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from distributed_reinforcement_learning_amd.agents.base import AgentBase, clip_rewards
from distributed_reinforcement_learning_amd.algorithms import a2c
from distributed_reinforcement_learning_amd.models import ActorCritic, VectorActorCritic
from distributed_reinforcement_learning_amd.ops import FusedAdam


class Agent(AgentBase):
    def __init__(self, input_shape, num_action: int, discount_factor: float,
                 start_learning_rate: float, end_learning_rate: float,
                 learning_frame: int, baseline_loss_coef: float,
                 entropy_coef: float, gradient_clip_norm: float,
                 reward_clipping: str, device: str = "cpu",
                 compute_dtype: torch.dtype = torch.bfloat16,
                 build_optimizer: bool = True, seed: Optional[int] = None):
        super().__init__(device=device, compute_dtype=compute_dtype)
        self.input_shape = tuple(input_shape)
        self.num_action = num_action
        self.discount_factor = discount_factor
        self.start_learning_rate = start_learning_rate
        self.end_learning_rate = end_learning_rate
        self.learning_frame = learning_frame
        self.baseline_loss_coef = baseline_loss_coef
        self.entropy_coef = entropy_coef
        self.gradient_clip_norm = gradient_clip_norm
        self.reward_clipping = reward_clipping
        if seed is not None:
            torch.manual_seed(seed)
        self.rng = np.random.default_rng(seed)

        cls = ActorCritic if len(self.input_shape) == 3 else VectorActorCritic
        self.model = self.finalize_model(cls(self.input_shape, num_action))
        self.optimizer = None
        if build_optimizer:
            self.optimizer = FusedAdam(self.model.parameters(),
                                       lr=start_learning_rate,
                                       clip_norm=gradient_clip_norm)

    def compute_a2c_losses(self, s, ns, pa, a, r, d):
        """Pure loss body over device tensors (shared by train and the
        graphed learner step): (pi, baseline, entropy, total).

        On GPU the whole post-network pipeline (softmax, 1-step-TD
        advantage, all three losses + backward) is the fused K6 kernel
        pair (ops/hip/a2c_loss.hip); the CPU composition below is the
        golden the GPU parity test compares against."""
        from distributed_reinforcement_learning_amd import ops as _ops
        if s.is_cuda and _ops.available():
            from distributed_reinforcement_learning_amd.ops.a2c_op import (
                fused_a2c_loss,
            )
            logits, value = self.model.logits_value(s, pa)
            with torch.no_grad():
                # next-state eval: prev_action for s' is the current
                # action (reference train_a3c.py feed: npa_ph = action)
                _, next_value = self.model.logits_value(ns, a)
            return fused_a2c_loss(
                logits, value.float(), next_value.float(), a, r, d,
                self.discount_factor, self.reward_clipping,
                self.baseline_loss_coef, self.entropy_coef)
        clipped_r = clip_rewards(r, self.reward_clipping)
        discounts = (~d).float() * self.discount_factor
        with self.autocast():
            policy, value = self.model(s, pa)
            # next-state eval: prev_action for s' is the current action
            # (reference train_a3c.py feed: npa_ph = action)
            _, next_value = self.model(ns, a)
        policy, value = policy.float(), value.float()
        next_value = next_value.float().detach()
        pi_loss = a2c.compute_policy_loss(policy, a, value, next_value,
                                          discounts, clipped_r)
        baseline_loss = a2c.compute_baseline_loss(value, next_value,
                                                  discounts, clipped_r)
        entropy = a2c.compute_entropy_loss(policy)
        total = (pi_loss + baseline_loss * self.baseline_loss_coef
                 + entropy * self.entropy_coef)
        return pi_loss, baseline_loss, entropy, total

    def train(self, state, next_state, previous_action, action, reward,
              done) -> Tuple[float, float, float, float]:
        s = self.frames_to_device(state)
        ns = self.frames_to_device(next_state)
        pa = self.to_device(previous_action, torch.int64)
        a = self.to_device(action, torch.int64)
        r = self.to_device(reward, torch.float32)
        d = self.to_device(done, torch.bool)

        pi_loss, baseline_loss, entropy, total = self.compute_a2c_losses(
            s, ns, pa, a, r, d)

        self.optimizer.zero_grad()
        total.backward()
        self.reduce_gradients()
        lr = self.lr_at(self.global_step)
        self.optimizer.step(lr=lr)
        self.global_step += 1
        self.num_env_frames += len(r)
        return (float(pi_loss.detach()), float(baseline_loss.detach()),
                float(entropy.detach()), lr)

    @torch.no_grad()
    def get_policy_and_action(self, state, previous_action):
        s = self.frames_to_device(np.asarray(state)[None])
        pa = self.to_device(np.asarray([previous_action]), torch.int64)
        policy, _ = self.model(s, pa)
        p = policy[0].float().cpu().numpy().astype(np.float64)
        p = p / p.sum()
        action = int(self.rng.choice(self.num_action, p=p))
        return action, p.astype(np.float32), float(p.max())
