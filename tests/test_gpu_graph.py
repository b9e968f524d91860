"""hipGraph-captured IMPALA step: parity vs the eager path. GPU-only."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu


def _agent(seed=0):
    from distributed_reinforcement_learning_amd.agents import impala
    return impala.Agent(
        trajectory=8, input_shape=[84, 84, 4], num_action=6,
        lstm_hidden_size=32, discount_factor=0.99, start_learning_rate=1e-3,
        end_learning_rate=0.0, learning_frame=10 ** 9,
        baseline_loss_coef=1.0, entropy_coef=0.05, gradient_clip_norm=40.0,
        reward_clipping="abs_one", device="cuda:0", seed=seed)


def _batch(B=4, T=8, A=6, H=32, seed=0):
    rng = np.random.default_rng(seed)
    return dict(
        state=rng.integers(0, 255, (B, T, 84, 84, 4), dtype=np.uint8),
        reward=rng.normal(size=(B, T)).astype(np.float32),
        action=rng.integers(0, A, (B, T)).astype(np.int32),
        done=np.zeros((B, T), dtype=bool),
        behavior_policy=np.full((B, T, A), 1 / A, dtype=np.float32),
        previous_action=rng.integers(0, A, (B, T)).astype(np.int32),
        initial_h=(rng.normal(size=(B, T, H)) * 0.1).astype(np.float32),
        initial_c=(rng.normal(size=(B, T, H)) * 0.1).astype(np.float32),
    )


def test_graphed_step_matches_eager():
    from distributed_reinforcement_learning_amd.runtime import GraphedImpalaStep
    torch.manual_seed(0)
    a_eager = _agent(seed=0)
    a_graph = _agent(seed=0)
    a_graph.model.load_state_dict(a_eager.model.state_dict())
    a_graph.optimizer.flat_params.copy_(a_eager.optimizer.flat_params)

    graphed = GraphedImpalaStep(a_graph, batch_size=4)
    # graph construction must not have perturbed weights
    assert torch.equal(a_graph.optimizer.flat_params,
                       a_eager.optimizer.flat_params)

    for i in range(3):
        b = _batch(seed=i)
        out_e = a_eager.train(**{
            "state": b["state"], "reward": b["reward"],
            "action": b["action"], "done": b["done"],
            "behavior_policy": b["behavior_policy"],
            "previous_action": b["previous_action"],
            "initial_h": b["initial_h"], "initial_c": b["initial_c"]})
        graphed.step(b)
        out_g = graphed.last_losses()
        # bf16 forward: small numeric differences are expected; losses and
        # resulting weights must agree closely
        assert out_g[0] == pytest.approx(out_e[0], rel=5e-2, abs=5e-1)
        assert out_g[3] == out_e[3]  # identical lr schedule
    assert torch.allclose(a_graph.optimizer.flat_params,
                          a_eager.optimizer.flat_params, atol=1e-2)


def test_graphed_step_updates_weights_every_replay():
    from distributed_reinforcement_learning_amd.runtime import GraphedImpalaStep
    agent = _agent(seed=1)
    graphed = GraphedImpalaStep(agent, batch_size=4)
    w0 = agent.optimizer.flat_params.detach().clone()
    graphed.step(_batch(seed=10))
    w1 = agent.optimizer.flat_params.detach().clone()
    graphed.step(_batch(seed=11))
    w2 = agent.optimizer.flat_params.detach().clone()
    assert not torch.equal(w0, w1)
    assert not torch.equal(w1, w2)
    assert agent.global_step == 2


def test_graphed_a3c_train_step():
    """GraphedTrainStep (the A3C GPU learner's captured loss+optimizer
    step, runtime/replay_graphed.py) over the fused K6 loss: capture must
    not perturb weights, replays must train, losses stay finite."""
    from distributed_reinforcement_learning_amd.agents import a3c
    from distributed_reinforcement_learning_amd.runtime.replay_graphed import (
        GraphedTrainStep,
    )
    rng = np.random.default_rng(4)
    N, A = 8, 4
    agent = a3c.Agent(
        input_shape=[84, 84, 4], num_action=A, discount_factor=0.997,
        baseline_loss_coef=1.0, entropy_coef=0.05,
        start_learning_rate=1e-3, end_learning_rate=0.0,
        learning_frame=10 ** 9, gradient_clip_norm=40.0,
        reward_clipping="abs_one", device="cuda:0", seed=2)

    def mk():
        return {
            "state": torch.as_tensor(rng.integers(
                0, 255, (N, 84, 84, 4), dtype=np.uint8)).cuda(),
            "next_state": torch.as_tensor(rng.integers(
                0, 255, (N, 84, 84, 4), dtype=np.uint8)).cuda(),
            "pa": torch.as_tensor(rng.integers(0, A, N)).cuda(),
            "a": torch.as_tensor(rng.integers(0, A, N)).cuda(),
            "r": torch.as_tensor(
                rng.normal(size=N).astype(np.float32)).cuda(),
            "d": torch.as_tensor(rng.random(N) < 0.1).cuda(),
        }

    def loss_fn(i):
        pi, bl, ent, total = agent.compute_a2c_losses(
            agent.frames_to_device(i["state"]),
            agent.frames_to_device(i["next_state"]),
            i["pa"].long(), i["a"].long(), i["r"], i["d"])
        return pi, bl, ent, total

    p_before = agent.optimizer.flat_params.detach().clone()
    g = GraphedTrainStep(agent, mk(), loss_fn)
    assert torch.equal(agent.optimizer.flat_params, p_before)
    for _ in range(4):
        out = g.step(mk())
        assert len(out) == 4
    torch.cuda.synchronize()
    assert all(np.isfinite(float(x)) for x in out[:3])
    assert not torch.equal(agent.optimizer.flat_params, p_before)
