"""World>1 RCCL-in-graph proof on a single GPU.

Runs scripts/dist_graph_check.py as 2 torchrun ranks BOTH pinned to
cuda:0: the captured gather -> RCCL all-reduce -> fused-update graph must
capture (the loud-failure policy in parallel/dist.py raises on fallback)
and keep rank-disjoint training bit-identical across ranks.
"""

import os
import subprocess
import sys
from pathlib import Path

import pytest
import torch

ROOT = Path(__file__).resolve().parent.parent

pytestmark = pytest.mark.gpu


@pytest.mark.skipif(not torch.cuda.is_available(), reason="needs a GPU")
def test_captured_rccl_allreduce_world2_one_gpu():
    env = dict(os.environ)
    env.setdefault("HSA_ENABLE_IPC_MODE_LEGACY", "0")
    env["MASTER_ADDR"] = "127.0.0.1"
    cmd = [sys.executable, "-m", "torch.distributed.run",
           "--nnodes=1", "--nproc-per-node", "2",
           "--master-addr", "127.0.0.1", "--master-port", "29537",
           str(ROOT / "scripts" / "dist_graph_check.py")]
    run = subprocess.run(cmd, cwd=str(ROOT), env=env, capture_output=True,
                         text=True, timeout=600)
    sys.stderr.write(run.stdout[-4000:])
    sys.stderr.write(run.stderr[-4000:])
    assert run.returncode == 0, "dist graph check failed (see output)"
    assert "DIST_GRAPH_OK" in run.stdout
