import numpy as np

from distributed_reinforcement_learning_amd.envs import (
    CartPoleEnv, SyntheticAtariEnv, make_env, make_uint8_env,
    pomdp_uint8_env,
)
from distributed_reinforcement_learning_amd.envs.wrappers import (
    ProcessFrame84, _AreaResizer,
)


def test_cartpole_episode():
    env = CartPoleEnv(seed=0)
    obs = env.reset()
    assert obs.shape == (4,)
    steps = 0
    done = False
    while not done and steps < 500:
        obs, r, done, _ = env.step(steps % 2)
        assert r == 1.0
        steps += 1
    assert done


def test_synthetic_env_frames():
    env = SyntheticAtariEnv(num_actions=18, seed=0, episode_len=5)
    f = env.reset()
    assert f.shape == (210, 160, 3) and f.dtype == np.uint8
    for i in range(5):
        f, r, done, _ = env.step(0)
    assert done


def test_process_frame84():
    frame = np.random.default_rng(0).integers(
        0, 256, size=(210, 160, 3), dtype=np.uint8)
    out = ProcessFrame84.process(frame)
    assert out.shape == (84, 84, 1) and out.dtype == np.uint8


def test_area_resizer_preserves_mean():
    """Box-filter resize preserves total mass (INTER_AREA property)."""
    rng = np.random.default_rng(1)
    img = rng.random((210, 160)).astype(np.float32)
    r = _AreaResizer(210, 160, 110, 84)
    out = r(img)
    assert out.shape == (110, 84)
    np.testing.assert_allclose(out.mean(), img.mean(), rtol=1e-5)


def test_area_resizer_constant_image():
    img = np.full((210, 160), 7.0, dtype=np.float32)
    out = _AreaResizer(210, 160, 110, 84)(img)
    np.testing.assert_allclose(out, 7.0, rtol=1e-6)


def test_uint8_env_pipeline():
    env = make_uint8_env("BreakoutDeterministic-v4", num_actions=4, seed=0)
    obs = env.reset()
    assert obs.shape == (84, 84, 4) and obs.dtype == np.uint8
    obs, r, done, _ = env.step(0)
    assert obs.shape == (84, 84, 4)
    # frame stack: newest frame occupies the last channel
    assert obs[..., -1].any()


def test_pomdp_env_blanks_frames():
    env = pomdp_uint8_env("BreakoutDeterministic-v4", num_actions=4, seed=0)
    obs = env.reset()
    assert obs.shape == (84, 84, 1)
    blanks = 0
    total = 200
    for _ in range(total):
        obs, _, done, _ = env.step(0)
        if not obs.any():
            blanks += 1
        if done:
            env.reset()
    # ~20% blanking probability (wrappers.py:96-101)
    assert 0.08 < blanks / total < 0.35


def test_make_env_cartpole_shortcut():
    env = make_env("CartPole-v0", seed=0)
    assert env.reset().shape == (4,)
