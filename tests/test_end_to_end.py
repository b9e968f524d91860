"""End-to-end spawn-mode runs of all four trainers on tiny configs:
learner + real actor processes + shm transport + weight sync + TB logging +
checkpointing, on CPU."""

import json
import multiprocessing as mp
import os

import numpy as np
import pytest


def _tiny_config(tmp_path):
    def block(env, n_actors, avail, **kw):
        d = {"server_ip": "127.0.0.1", "server_port": 8123,
             "num_actors": n_actors, "env": [env] * n_actors,
             "available_action": [avail] * n_actors,
             "start_learning_rate": 1e-3, "end_learning_rate": 0.0,
             "learning_frame": 10 ** 9, "gradient_clip_norm": 40.0,
             "baseline_loss_coef": 1.0, "entropy_coef": 0.01,
             "discount_factor": 0.99, "reward_clipping": "abs_one"}
        d.update(kw)
        return d

    cfg = {
        "a3c_cartpole": block("CartPole-v0", 2, 2, model_input=[4],
                              model_output=2, queue_size=16, batch_size=4,
                              trajectory=8, reward_clipping="none"),
        "impala": block("SyntheticAtari", 2, 4, model_input=[84, 84, 4],
                        model_output=4, queue_size=16, batch_size=2,
                        trajectory=6, lstm_size=8),
        # vectorized actors: 1 process x 4 envs, batched inference
        "impala_vec": block("SyntheticAtari", 1, 4,
                            model_input=[84, 84, 4], model_output=4,
                            queue_size=16, batch_size=2, trajectory=6,
                            lstm_size=8, envs_per_actor=4),
        "apex": block("SyntheticAtari", 1, 4, model_input=[84, 84, 4],
                      model_output=4, queue_size=8, batch_size=4,
                      trajectory=4),
        "apex_vec": block("SyntheticAtari", 1, 4, model_input=[84, 84, 4],
                          model_output=4, queue_size=8, batch_size=4,
                          trajectory=4, envs_per_actor=3),
        "r2d2": block("SyntheticAtari", 1, 4, model_input=[84, 84, 1],
                      model_output=4, queue_size=8, batch_size=2,
                      seq_len=5, burn_in=2, lstm_size=8),
        "r2d2_vec": block("SyntheticAtari", 1, 4, model_input=[84, 84, 1],
                          model_output=4, queue_size=8, batch_size=2,
                          seq_len=5, burn_in=2, lstm_size=8,
                          envs_per_actor=3),
    }
    p = tmp_path / "config.json"
    p.write_text(json.dumps(cfg))
    return str(p)


def _run_main(algo, block, cfg_path, max_steps, q):
    try:
        os.environ.pop("WORLD_SIZE", None)
        import importlib
        mod = importlib.import_module(
            f"distributed_reinforcement_learning_amd.trainers.{algo}")
        mod.main(["--spawn", "--config", cfg_path,
                  "--algorithm_block", block,
                  "--max_steps", str(max_steps), "--seed", "0",
                  "--checkpoint_every", str(max_steps),
                  "--publish_every", "1"])
        q.put("ok")
    except Exception as e:  # pragma: no cover
        import traceback
        q.put("fail: " + traceback.format_exc())


@pytest.mark.parametrize("algo,block,max_steps,timeout", [
    ("a3c", "a3c_cartpole", 4, 120),
    ("impala", "impala", 3, 240),
    ("impala", "impala_vec", 3, 240),
    ("apex", "apex_vec", 3, 300),
    ("r2d2", "r2d2_vec", 3, 300),
    ("apex", "apex", 3, 300),
    ("r2d2", "r2d2", 3, 300),
])
def test_trainer_end_to_end(tmp_path, algo, block, max_steps, timeout):
    cfg_path = _tiny_config(tmp_path)
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        # spawn (not fork): the pytest process has already run parallel torch
        # ops, and a forked learner would deadlock in the inherited OpenMP
        # pool on its first at::parallel_for.
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        p = ctx.Process(target=_run_main,
                        args=(algo, block, cfg_path, max_steps, q))
        p.start()
        try:
            result = q.get(timeout=timeout)
        finally:
            p.join(timeout=30)
            if p.is_alive():
                p.terminate()
                p.join()
        assert result == "ok", result
        # learner TB run dir + scalars exist (reference layout: runs/learner)
        scalars = tmp_path / "runs" / "learner" / "scalars.jsonl"
        assert scalars.exists()
        tags = {json.loads(l)["tag"] for l in scalars.read_text().splitlines()}
        expected = {"data/time"} | (
            {"data/pi_loss", "data/value_loss", "data/entropy", "data/lr"}
            if algo in ("a3c", "impala") else {"data/loss"})
        assert expected <= tags, f"missing scalars: {expected - tags}"
        # checkpoint written with the standard layout
        ck = tmp_path / "checkpoints" / f"{block}.pt"
        assert ck.exists()
        import torch
        blob = torch.load(str(ck), map_location="cpu", weights_only=False)
        assert {"model", "optimizer", "global_step"} <= set(blob)
        assert blob["global_step"] == max_steps
    finally:
        os.chdir(cwd)


def test_evaluate_script_on_trained_checkpoint(tmp_path):
    """scripts/evaluate.py loads a checkpoint written by a spawn run and
    reports episode scores (standalone evaluation entrypoint)."""
    import subprocess
    import sys
    cfg_path = _tiny_config(tmp_path)
    cwd = os.getcwd()
    os.chdir(tmp_path)
    try:
        ctx = mp.get_context("spawn")
        q = ctx.Queue()
        p = ctx.Process(target=_run_main,
                        args=("a3c", "a3c_cartpole", cfg_path, 3, q))
        p.start()
        result = q.get(timeout=120)
        p.join(timeout=30)
        assert result == "ok", result
        repo = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))
        run = subprocess.run(
            [sys.executable, os.path.join(repo, "scripts", "evaluate.py"),
             "--algorithm", "a3c", "--algorithm_block", "a3c_cartpole",
             "--checkpoint", str(tmp_path / "checkpoints"
                                 / "a3c_cartpole.pt"),
             "--config", cfg_path, "--episodes", "2"],
            capture_output=True, text=True, timeout=120)
        assert run.returncode == 0, run.stderr[-2000:]
        out = json.loads(run.stdout.strip().splitlines()[-1])
        assert out["episodes"] == 2 and out["mean_episode_steps"] > 0
    finally:
        os.chdir(cwd)
