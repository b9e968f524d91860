"""Synthetic Atari-shaped env for offline benchmarking.

Emits raw 210x160x3 uint8 frames (what a real ALE env emits before the
preprocessing pipeline) with a cheap procedural pattern, a Bernoulli reward and
geometric episode lengths, so the full wrapper stack and actor loop run at
realistic shapes with zero external dependencies. BASELINE.json mandates
synthetic frames for all measured configs (no network for ROMs).
"""

from __future__ import annotations

from typing import Optional

import numpy as np

from distributed_reinforcement_learning_amd.envs.base import Env


class SyntheticAtariEnv(Env):
    observation_shape = (210, 160, 3)

    def __init__(self, num_actions: int = 18, seed: Optional[int] = None,
                 episode_len: int = 500, reward_p: float = 0.05,
                 height: int = 210, width: int = 160):
        self.action_space_n = num_actions
        self.rng = np.random.default_rng(seed)
        self.episode_len = episode_len
        self.reward_p = reward_p
        self.h, self.w = height, width
        self.observation_shape = (height, width, 3)
        self._t = 0
        # pre-generate a small bank of frames; per-step we roll + add noise so
        # frames vary without a full 100KB RNG draw per step.
        self._bank = self.rng.integers(
            0, 256, size=(8, self.h, self.w, 3), dtype=np.uint8)

    def get_action_meanings(self):
        base = ["NOOP", "FIRE", "RIGHT", "LEFT"]
        return (base + [f"A{i}" for i in range(4, self.action_space_n)])[
            : self.action_space_n]

    def _frame(self) -> np.ndarray:
        f = self._bank[self._t % len(self._bank)]
        return np.roll(f, shift=self._t % self.w, axis=1)

    def reset(self) -> np.ndarray:
        self._t = 0
        return self._frame()

    def step(self, action: int):
        self._t += 1
        reward = float(self.rng.random() < self.reward_p)
        done = self._t >= self.episode_len
        return self._frame(), reward, done, {}
