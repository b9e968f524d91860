"""Multi-rank trainer topology on CPU: 2 learner ranks (torchrun, gloo)
with spawned actors — trajectory scatter across rank-owned ring shards,
rank-0 init broadcast, per-step gradient all-reduce, rank-0 weight
publish/checkpoint. The same code path RCCL takes on a multi-GPU node
(the driver's SCALE run), minus the hipGraph capture."""

import json
import os
import subprocess
import sys

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))


def test_impala_two_rank_learner_spawn(tmp_path):
    cfg = {
        "impala": {
            "server_ip": "127.0.0.1", "server_port": 8231,
            "num_actors": 2,
            "env": ["SyntheticAtari"] * 2,
            "available_action": [4] * 2,
            "start_learning_rate": 1e-3, "end_learning_rate": 0.0,
            "learning_frame": 10 ** 9, "gradient_clip_norm": 40.0,
            "baseline_loss_coef": 1.0, "entropy_coef": 0.01,
            "discount_factor": 0.99, "reward_clipping": "abs_one",
            "model_input": [84, 84, 4], "model_output": 4,
            "queue_size": 16, "batch_size": 2, "trajectory": 6,
            "lstm_size": 8,
        }
    }
    cfg_path = tmp_path / "config.json"
    cfg_path.write_text(json.dumps(cfg))
    env = dict(os.environ)
    for var in ("WORLD_SIZE", "RANK", "LOCAL_RANK", "MASTER_PORT"):
        env.pop(var, None)
    env["MASTER_ADDR"] = "127.0.0.1"
    env["CUDA_VISIBLE_DEVICES"] = ""  # force the gloo/CPU path
    run = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29541", os.path.join(REPO, "train_impala.py"),
         "--spawn", "--config", str(cfg_path), "--max_steps", "3",
         "--seed", "0", "--publish_every", "1",
         "--checkpoint_every", "3"],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=420)
    sys.stderr.write(run.stdout[-3000:])
    sys.stderr.write(run.stderr[-3000:])
    assert run.returncode == 0
    # rank 0 wrote the learner scalars + checkpoint; rank 1 trained its
    # own shard (its log dir exists)
    assert (tmp_path / "runs" / "learner" / "scalars.jsonl").exists()
    assert (tmp_path / "runs" / "learner_rank1").exists()
    assert (tmp_path / "checkpoints" / "impala.pt").exists()


def test_capture_failure_policy(monkeypatch):
    """handle_capture_failure: FAIL-FAST by default (re-raises), continue
    with a loud warning only under DRLA_ALLOW_EAGER_REDUCE=1."""
    import pytest
    from distributed_reinforcement_learning_amd.parallel.dist import (
        handle_capture_failure,
    )
    err = RuntimeError("capture refused")
    monkeypatch.delenv("DRLA_ALLOW_EAGER_REDUCE", raising=False)
    try:
        raise err
    except RuntimeError:
        with pytest.raises(RuntimeError):
            handle_capture_failure(err)
    monkeypatch.setenv("DRLA_ALLOW_EAGER_REDUCE", "1")
    try:
        raise err
    except RuntimeError:
        handle_capture_failure(err)  # must not raise
