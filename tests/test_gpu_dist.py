"""Captured-RCCL-in-hipGraph proof.

Adaptive to the box: with >=2 visible devices the check runs 2 torchrun
ranks on cuda:0/cuda:1 (the full world>1 proof — rank-disjoint training
must stay bit-identical through the captured all-reduce); on a 1-GPU box
it runs a REAL 1-rank RCCL communicator with DRLA_FORCE_DIST_GRAPH=1,
proving RCCL collectives capture and replay inside a hipGraph on this
stack. Two ranks on ONE device is impossible at the RCCL layer: 2.26.6
hard-rejects it ("Duplicate GPU detected", ncclInvalidUsage — measured)
and the pool blocks CPX compute partitioning. Capture failure at world>1
FAIL-FASTS (parallel/dist.py handle_capture_failure), so the eager
fallback can no longer mask a broken multi-rank graph.
"""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

ROOT = Path(__file__).resolve().parent.parent

pytestmark = pytest.mark.gpu


def _run_check(nproc: int) -> str:
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["MASTER_ADDR"] = "127.0.0.1"
    env.pop("DRLA_ALLOW_EAGER_REDUCE", None)  # strict: capture or fail
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", str(nproc),
           "--master-addr", "127.0.0.1", "--master-port", "29537",
           str(ROOT / "scripts" / "dist_graph_check.py")]
    run = subprocess.run(cmd, cwd=str(ROOT), env=env, capture_output=True,
                         text=True, timeout=600)
    sys.stderr.write(run.stdout[-4000:])
    sys.stderr.write(run.stderr[-4000:])
    assert run.returncode == 0, "dist graph check failed (see output)"
    return run.stdout


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs a GPU")
def test_captured_rccl_allreduce_in_graph():
    n = min(2, torch.cuda.device_count())
    out = _run_check(n)
    assert f"DIST_GRAPH_OK world={n}" in out
    if n > 1:
        assert "params_equal=True" in out
