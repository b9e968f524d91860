"""CartPole-v0 physics, dependency-free.

Standard cart-pole dynamics (Barto-Sutton-Anderson; same constants gym uses)
so the A3C plumbing config (BASELINE config #1) runs without gym. Obs is the
4-vector [x, x_dot, theta, theta_dot]; episode ends on |x|>2.4, |theta|>12deg,
or 200 steps; reward 1 per step.
"""

from __future__ import annotations

import math
from typing import Optional

import numpy as np

from distributed_reinforcement_learning_amd.envs.base import Env


class CartPoleEnv(Env):
    action_space_n = 2
    observation_shape = (4,)

    def __init__(self, seed: Optional[int] = None, max_steps: int = 200):
        self.rng = np.random.default_rng(seed)
        self.max_steps = max_steps
        self.gravity = 9.8
        self.masscart = 1.0
        self.masspole = 0.1
        self.total_mass = self.masscart + self.masspole
        self.length = 0.5
        self.polemass_length = self.masspole * self.length
        self.force_mag = 10.0
        self.tau = 0.02
        self.theta_threshold = 12 * 2 * math.pi / 360
        self.x_threshold = 2.4
        self.state = None
        self._steps = 0

    def reset(self) -> np.ndarray:
        self.state = self.rng.uniform(-0.05, 0.05, size=4).astype(np.float32)
        self._steps = 0
        return self.state.copy()

    def step(self, action: int):
        x, x_dot, theta, theta_dot = self.state
        force = self.force_mag if action == 1 else -self.force_mag
        costheta, sintheta = math.cos(theta), math.sin(theta)
        temp = (force + self.polemass_length * theta_dot ** 2 * sintheta) \
            / self.total_mass
        thetaacc = (self.gravity * sintheta - costheta * temp) / (
            self.length * (4.0 / 3.0
                           - self.masspole * costheta ** 2 / self.total_mass))
        xacc = temp - self.polemass_length * thetaacc * costheta \
            / self.total_mass
        x = x + self.tau * x_dot
        x_dot = x_dot + self.tau * xacc
        theta = theta + self.tau * theta_dot
        theta_dot = theta_dot + self.tau * thetaacc
        self.state = np.array([x, x_dot, theta, theta_dot], dtype=np.float32)
        self._steps += 1
        done = bool(
            abs(x) > self.x_threshold
            or abs(theta) > self.theta_threshold
            or self._steps >= self.max_steps)
        return self.state.copy(), 1.0, done, {}
