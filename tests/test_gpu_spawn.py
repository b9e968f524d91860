"""Full-system spawn smoke ON the GPU: hipGraph-captured IMPALA learner +
2 vectorized actor processes + shm rings + seqlock weight publish, end to
end. (The CPU e2e suite runs the same topology eagerly; this is the
GPU-learner variant the production path actually uses.)"""

import json
import os
import subprocess
import sys

import pytest
import torch

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

pytestmark = pytest.mark.gpu


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs a GPU")
def test_impala_gpu_spawn_smoke(tmp_path):
    cfg = {
        "impala": {
            "server_ip": "127.0.0.1", "server_port": 8337,
            "num_actors": 2,
            "env": ["SyntheticAtari"] * 2,
            "available_action": [4] * 2,
            "start_learning_rate": 1e-3, "end_learning_rate": 0.0,
            "learning_frame": 10 ** 9, "gradient_clip_norm": 40.0,
            "baseline_loss_coef": 1.0, "entropy_coef": 0.01,
            "discount_factor": 0.99, "reward_clipping": "abs_one",
            "model_input": [84, 84, 4], "model_output": 4,
            "queue_size": 16, "batch_size": 4, "trajectory": 8,
            "lstm_size": 64, "envs_per_actor": 2,
        }
    }
    cfg_path = tmp_path / "config.json"
    cfg_path.write_text(json.dumps(cfg))
    env = dict(os.environ)
    for var in ("WORLD_SIZE", "RANK", "LOCAL_RANK"):
        env.pop(var, None)
    run = subprocess.run(
        [sys.executable, os.path.join(REPO, "train_impala.py"), "--spawn",
         "--config", str(cfg_path), "--max_steps", "8", "--seed", "0",
         "--publish_every", "2", "--checkpoint_every", "8"],
        cwd=str(tmp_path), env=env, capture_output=True, text=True,
        timeout=420)
    sys.stderr.write(run.stdout[-2000:])
    sys.stderr.write(run.stderr[-2000:])
    assert run.returncode == 0
    scalars = tmp_path / "runs" / "learner" / "scalars.jsonl"
    assert scalars.exists()
    lines = [json.loads(l) for l in scalars.read_text().splitlines()]
    losses = [l for l in lines if l["tag"] == "data/pi_loss"]
    assert losses and all(abs(l["value"]) < 1e6 for l in losses)
    assert (tmp_path / "checkpoints" / "impala.pt").exists()
