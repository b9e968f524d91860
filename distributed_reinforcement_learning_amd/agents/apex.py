"""Ape-X DQN agent — capability-parity with reference agent/apex.py:
dueling double-DQN with a target network, PER-weighted updates
(distributed_train, agent/apex.py:136-153), TD-error scoring for priority
init (get_td_error, :119-134), epsilon-greedy acting (:92-107), and
target_to_main hard sync (:82).

Note the reference's target-net quirk kept for parity: the next-state evals
feed the *current action* as the previous-action embedding for s'
(apex_value.py:44-63).
"""

from __future__ import annotations

from typing import Optional, Tuple

import numpy as np
import torch

from distributed_reinforcement_learning_amd.agents.base import AgentBase, clip_rewards
from distributed_reinforcement_learning_amd.algorithms import dqn
from distributed_reinforcement_learning_amd.models import ApexDuelingQ, VectorDuelingQ
from distributed_reinforcement_learning_amd.ops import FusedAdam


class Agent(AgentBase):
    def __init__(self, input_shape, num_action: int, discount_factor: float,
                 gradient_clip_norm: float, reward_clipping: str,
                 start_learning_rate: float, end_learning_rate: float,
                 learning_frame: int, device: str = "cpu",
                 compute_dtype: torch.dtype = torch.bfloat16,
                 build_optimizer: bool = True, seed: Optional[int] = None):
        super().__init__(device=device, compute_dtype=compute_dtype)
        self.input_shape = tuple(input_shape)
        self.num_action = num_action
        self.discount_factor = discount_factor
        self.gradient_clip_norm = gradient_clip_norm
        self.reward_clipping = reward_clipping
        self.start_learning_rate = start_learning_rate
        self.end_learning_rate = end_learning_rate
        self.learning_frame = learning_frame
        if seed is not None:
            torch.manual_seed(seed)
        self.rng = np.random.default_rng(seed)

        cls = ApexDuelingQ if len(self.input_shape) == 3 else VectorDuelingQ
        self.model = self.finalize_model(cls(self.input_shape, num_action))
        self.target_model = self.finalize_model(
            cls(self.input_shape, num_action))
        self.target_model.load_state_dict(self.model.state_dict())
        for p in self.target_model.parameters():
            p.requires_grad_(False)
        self.optimizer = None
        if build_optimizer:
            self.optimizer = FusedAdam(self.model.parameters(),
                                       lr=start_learning_rate,
                                       clip_norm=gradient_clip_norm)

    # -- target sync (reference utils.py:27-32 grouped assigns) --------------

    def target_to_main(self) -> None:
        self.target_model.load_state_dict(self.model.state_dict())

    main_to_target = target_to_main  # both directions are a hard copy here

    def _checkpoint_extra(self):
        return {"target_model": self.target_model.state_dict()}

    def _load_checkpoint_extra(self, blob):
        if "target_model" in blob:
            self.target_model.load_state_dict(blob["target_model"])

    # -- forward helpers -----------------------------------------------------

    def _targets(self, s, ns, pa, a, r, d):
        clipped_r = clip_rewards(r, self.reward_clipping)
        discounts = (~d).float() * self.discount_factor
        with self.autocast():
            main_q = self.model(s, pa).float()
            with torch.no_grad():
                next_main_q = self.model(ns, a).float()
                next_target_q = self.target_model(ns, a).float()
        target_value, _ = dqn.double_dqn_target(next_main_q, next_target_q,
                                                clipped_r, discounts)
        sav = dqn.take_state_action_value(main_q, a)
        return target_value, sav

    def _prep(self, state, next_state, previous_action, action, reward, done):
        return (self.frames_to_device(state),
                self.frames_to_device(next_state),
                self.to_device(previous_action, torch.int64),
                self.to_device(action, torch.int64),
                self.to_device(reward, torch.float32),
                self.to_device(done, torch.bool))

    # -- public API ----------------------------------------------------------

    @torch.no_grad()
    def get_td_error(self, state, next_state, previous_action, action,
                     reward, done, as_tensor: bool = False):
        s, ns, pa, a, r, d = self._prep(state, next_state, previous_action,
                                        action, reward, done)
        target_value, sav = self._targets(s, ns, pa, a, r, d)
        td = (target_value - sav).abs()
        return td if as_tensor else td.cpu().numpy()

    def compute_distributed_loss(self, s, ns, pa, a, r, d, w):
        """Pure GPU loss body (device tensors in, no optimizer/python
        state) — shared by the eager distributed_train and the hipGraph-
        captured replay step (runtime/replay_graphed.py).

        Fused K8 path: double-DQN target + IS-weighted TD loss and its
        closed-form backward in one kernel each (ops/hip/dqn_loss.hip);
        ONE main-net forward over [s ; ns] (the no-grad next-state eval
        rides in the same launches)."""
        from distributed_reinforcement_learning_amd.ops.dqn_op import (
            fused_dqn_loss,
        )
        clipped_r = clip_rewards(r, self.reward_clipping)
        discounts = (~d).float() * self.discount_factor
        B = s.shape[0]
        # target-net eval is independent of the main forward — overlap it
        # on a side stream (capture-legal fork/join, same pattern as the
        # R2D2 agent's _window_qs)
        side = getattr(self, "_tgt_stream", None)
        if side is None and self.device.type == "cuda":
            side = self._tgt_stream = torch.cuda.Stream()
        if side is not None:
            main_stream = torch.cuda.current_stream()
            side.wait_stream(main_stream)
            with torch.cuda.stream(side), torch.no_grad():
                next_target_q = self.target_model(ns, a)
        else:
            with torch.no_grad():
                next_target_q = self.target_model(ns, a)
        qs = self.model(torch.cat([s, ns]), torch.cat([pa, a]))
        main_q = qs[:B]
        next_main_q = qs[B:].detach()
        if side is not None:
            main_stream.wait_stream(side)
            if not torch.cuda.is_current_stream_capturing():
                next_target_q.record_stream(main_stream)
        return fused_dqn_loss(main_q, next_main_q, next_target_q, a,
                              clipped_r, discounts, w)

    def distributed_train(self, state, next_state, previous_action, action,
                          reward, done, is_weight, as_tensor: bool = False
                          ) -> Tuple[float, np.ndarray]:
        """PER-weighted update; returns (loss, |td_error|) for priority
        refresh (reference agent/apex.py:136-153)."""
        s, ns, pa, a, r, d = self._prep(state, next_state, previous_action,
                                        action, reward, done)
        w = self.to_device(is_weight, torch.float32)
        if self.device.type == "cuda":
            loss, td_signed = self.compute_distributed_loss(s, ns, pa, a, r,
                                                            d, w)
            self.optimizer.zero_grad()
            loss.backward()
            self.reduce_gradients()
            lr = self.lr_at(self.global_step)
            self.optimizer.step(lr=lr)
            self.global_step += 1
            self.num_env_frames += len(r)
            td_error = td_signed.detach().abs()
            if not as_tensor:
                return float(loss.detach()), td_error.cpu().numpy()
            # as_tensor path: no host sync — callers float() the loss at
            # logging cadence only
            return loss.detach(), td_error
        target_value, sav = self._targets(s, ns, pa, a, r, d)
        td_sq = (target_value.detach() - sav) ** 2
        loss = (td_sq * w).mean()

        self.optimizer.zero_grad()
        loss.backward()
        self.reduce_gradients()
        lr = self.lr_at(self.global_step)
        self.optimizer.step(lr=lr)
        self.global_step += 1
        self.num_env_frames += len(r)
        td_error = (target_value - sav).detach().abs()
        if not as_tensor:
            td_error = td_error.cpu().numpy()
        return float(loss.detach()), td_error

    def train(self, state, next_state, previous_action, action, reward,
              done) -> Tuple[float, np.ndarray]:
        ones = np.ones(np.shape(reward), dtype=np.float32)
        return self.distributed_train(state, next_state, previous_action,
                                      action, reward, done, ones)

    @torch.no_grad()
    def target_main_test(self, state, previous_action):
        """Debug helper (reference agent/apex.py:109-117): print the
        next-state main/target Q values for one state."""
        s = self.frames_to_device(np.asarray(state)[None])
        a = self.to_device(np.asarray([previous_action]), torch.int64)
        next_main = self.model(s, a).float().cpu().numpy()
        target = self.target_model(s, a).float().cpu().numpy()
        print(next_main)
        print(target)
        return next_main, target

    @torch.no_grad()
    def get_actions_batch(self, states, previous_actions, epsilons):
        """Vectorized ε-greedy over E envs in ONE forward (vector actors,
        trainers/apex.py). Returns (actions [E], q [E,A], q_a [E])."""
        s = self.frames_to_device(np.asarray(states))
        pa = self.to_device(np.asarray(previous_actions), torch.int64)
        q = self.model(s, pa).float().cpu().numpy()
        E = len(q)
        greedy = q.argmax(axis=1)
        rand = self.rng.integers(self.num_action, size=E)
        explore = self.rng.random(E) <= np.asarray(epsilons)
        actions = np.where(explore, rand, greedy).astype(np.int64)
        return actions, q, q[np.arange(E), actions].astype(np.float32)

    @torch.no_grad()
    def get_policy_and_action(self, state, previous_action, epsilon: float
                              ) -> Tuple[int, np.ndarray, float]:
        s = self.frames_to_device(np.asarray(state)[None])
        pa = self.to_device(np.asarray([previous_action]), torch.int64)
        q = self.model(s, pa)[0].float().cpu().numpy()
        if self.rng.random() > epsilon:
            action = int(q.argmax())
        else:
            action = int(self.rng.integers(self.num_action))
        return action, q, float(q[action])
