"""bench.py driver-contract tests: single-process CPU run and the torchrun
multi-rank launch (gloo on CPU — the same code path RCCL takes on GPUs)."""

import json
import os
import subprocess
import sys

import pytest

REPO = os.path.dirname(os.path.dirname(os.path.abspath(__file__)))

BENCH_ARGS = ["--steps", "2", "--warmup", "1", "--batch", "2",
              "--trajectory", "4", "--lstm", "16", "--pool", "2"]


def _check_line(line: str, n_gpus: int):
    out = json.loads(line)
    assert out["metric"].startswith("learner env-frames/sec")
    assert out["n_gpus"] == n_gpus
    assert out["steps"] == 2 and out["warmup"] == 1
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["higher_is_better"] is True
    assert out["scaling"] == "weak"
    assert out["data"] == "synthetic"
    assert out["config"]["global_batch"] == 2 * n_gpus
    assert out["config"]["parallelism"] == f"dp{n_gpus}"
    return out


def test_bench_single_process():
    r = subprocess.run([sys.executable, "bench.py"] + BENCH_ARGS,
                       capture_output=True, text=True, cwd=REPO,
                       timeout=300)
    assert r.returncode == 0, r.stderr[-2000:]
    _check_line(r.stdout.strip().splitlines()[-1], 1)


@pytest.mark.parametrize("algo,extra", [
    ("apex", ["--batch", "4"]),
    ("r2d2", ["--batch", "2", "--seq-len", "6", "--burn-in", "2"]),
])
def test_bench_algo_flags(algo, extra):
    """--algo apex/r2d2 (BASELINE configs #3/#4) emit well-formed contract
    lines; CPU fallback exercises the same sample->train->priority-update
    loop against the float64 Memory."""
    r = subprocess.run(
        [sys.executable, "bench.py", "--algo", algo, "--steps", "2",
         "--warmup", "1"] + extra,
        capture_output=True, text=True, cwd=REPO, timeout=600)
    assert r.returncode == 0, r.stderr[-2000:]
    out = json.loads(r.stdout.strip().splitlines()[-1])
    assert algo in out["metric"].lower().replace("-", "")
    assert out["value"] > 0 and out["ms_per_step"] > 0
    assert out["data"] == "synthetic"
    if algo == "r2d2":
        assert out["config"]["seq_len"] == 6
        assert out["config"]["burn_in"] == 2


def test_bench_torchrun_two_ranks():
    """Exactly the driver's multi-GPU launch shape, on CPU/gloo."""
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    env.pop("RANK", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "2", "--master-addr", "127.0.0.1",
         "--master-port", "29511", "bench.py"] + BENCH_ARGS,
        capture_output=True, text=True, cwd=REPO, timeout=600, env=env)
    assert r.returncode == 0, r.stderr[-3000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1, "exactly one JSON line from rank 0"
    out = _check_line(json_lines[-1], 2)
    # whole-job aggregate: 2 ranks x 2 batch x 4 steps... value is frames/s;
    # just confirm the frames accounting doubled via global_batch above
    assert out["config"]["frames_per_step_per_rank"] == 8


def test_bench_torchrun_four_ranks():
    """The driver's N=4 launch shape (gloo on CPU): rank logic, pool
    seeding, MAX-over-ranks timing and the single rank-0 JSON line must
    hold beyond 2 ranks."""
    env = dict(os.environ)
    env.pop("WORLD_SIZE", None)
    env.pop("RANK", None)
    r = subprocess.run(
        [sys.executable, "-m", "torch.distributed.run", "--nnodes=1",
         "--nproc-per-node", "4", "--master-addr", "127.0.0.1",
         "--master-port", "29513", "bench.py"] + BENCH_ARGS,
        capture_output=True, text=True, cwd=REPO, timeout=600, env=env)
    assert r.returncode == 0, r.stderr[-3000:]
    json_lines = [l for l in r.stdout.splitlines() if l.startswith("{")]
    assert len(json_lines) == 1
    out = _check_line(json_lines[-1], 4)
    assert out["config"]["global_batch"] == 8
