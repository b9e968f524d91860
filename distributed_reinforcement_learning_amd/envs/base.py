"""Minimal env API (gym is not installable in this offline image).

Interface matches classic gym (`reset() -> obs`,
`step(a) -> (obs, reward, done, info)`) so the actor loops read like the
reference's (train_impala.py:132-194) and a real gym env can be dropped in via
``GymAdapter`` when gym is present.
"""

from __future__ import annotations

from typing import Any, Dict, Tuple

import numpy as np


class Env:
    action_space_n: int = 0
    observation_shape: Tuple[int, ...] = ()

    def reset(self) -> np.ndarray:
        raise NotImplementedError

    def step(self, action: int) -> Tuple[np.ndarray, float, bool, Dict[str, Any]]:
        raise NotImplementedError

    def get_action_meanings(self):
        return []

    @property
    def unwrapped(self) -> "Env":
        return self


class Wrapper(Env):
    def __init__(self, env: Env):
        self.env = env
        self.action_space_n = env.action_space_n
        self.observation_shape = env.observation_shape

    def reset(self):
        return self.env.reset()

    def step(self, action):
        return self.env.step(action)

    def get_action_meanings(self):
        return self.env.get_action_meanings()

    @property
    def unwrapped(self) -> Env:
        return self.env.unwrapped


class ObservationWrapper(Wrapper):
    def observation(self, obs: np.ndarray) -> np.ndarray:
        raise NotImplementedError

    def reset(self):
        return self.observation(self.env.reset())

    def step(self, action):
        obs, r, d, info = self.env.step(action)
        return self.observation(obs), r, d, info


class GymAdapter(Env):
    """Wrap a real gym env (old 4-tuple or new 5-tuple API) if gym exists.

    Translates ALE's life counter into the ``life_lost`` flag the trainer
    loops read (trainers/*.py ``info.get("life_lost")``): the reference
    actors track ``info['ale.lives']`` across steps and shape
    reward=-1/done=True on a decrease (/root/reference/train_impala.py:
    147-154); this adapter performs the tracking so every env — real ALE
    or synthetic — exposes the same flag."""

    def __init__(self, gym_env):
        self._env = gym_env
        self.action_space_n = gym_env.action_space.n
        self.observation_shape = tuple(gym_env.observation_space.shape)
        self._lives = None

    def reset(self):
        out = self._env.reset()
        self._lives = None
        return out[0] if isinstance(out, tuple) else out

    def step(self, action):
        out = self._env.step(action)
        if len(out) == 5:  # gymnasium API
            obs, r, term, trunc, info = out
            done = term or trunc
        else:
            obs, r, done, info = out
        lives = info.get("ale.lives", info.get("lives"))
        if lives is not None:
            if self._lives is not None and lives < self._lives:
                info = dict(info)
                info["life_lost"] = True
            self._lives = lives
        return obs, r, done, info

    def get_action_meanings(self):
        try:
            return self._env.unwrapped.get_action_meanings()
        except Exception:
            return []
