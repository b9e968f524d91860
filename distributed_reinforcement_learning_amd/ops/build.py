"""In-tree build of the _drla_hip extension with hipcc (gfx950 only).

Deliberately NOT torch.utils.cpp_extension's JIT path: the .so must live
in-tree (it travels to the GPU box with the repo snapshot; a JIT cache under
~/.cache does not), and the sources are native HIP — no hipify pass.

Usage:
    python -m distributed_reinforcement_learning_amd.ops.build
or via `python setup.py build_ext --inplace` / __graft_entry__.build().
"""

from __future__ import annotations

import os
import subprocess
import sys
import sysconfig
from typing import List

ARCH = os.environ.get("PYTORCH_ROCM_ARCH", "gfx950")

HERE = os.path.dirname(os.path.abspath(__file__))
HIP_DIR = os.path.join(HERE, "hip")
OUT_SO = os.path.join(HERE, "_drla_hip.so")

SOURCES = [
    "bind.cpp",
    "elementwise.hip",
    "vtrace.hip",
    "vtrace_loss.hip",
    "lstm_gates.hip",
    "optim.hip",
    "conv.hip",
    "per_tree.hip",
    "dqn_loss.hip",
    "a2c_loss.hip",
    "r2d2_loss.hip",
    "mlp_heads.hip",
]


def _existing_sources() -> List[str]:
    return [os.path.join(HIP_DIR, s) for s in SOURCES
            if os.path.exists(os.path.join(HIP_DIR, s))]


def build(verbose: bool = True, force: bool = False) -> str:
    import torch

    torch_dir = os.path.dirname(os.path.abspath(torch.__file__))
    sources = _existing_sources()
    if not force and os.path.exists(OUT_SO):
        so_mtime = os.path.getmtime(OUT_SO)
        if all(os.path.getmtime(s) < so_mtime for s in sources):
            if verbose:
                print(f"[drla build] {OUT_SO} up to date")
            return OUT_SO

    abi = "1" if torch.compiled_with_cxx11_abi() else "0"
    hipcc = os.environ.get("HIPCC", "hipcc")
    cmd = [
        hipcc, f"--offload-arch={ARCH}", "-O3", "-std=c++17", "-fPIC",
        "-shared",
        "-DTORCH_EXTENSION_NAME=_drla_hip",
        "-DTORCH_API_INCLUDE_EXTENSION_H",
        f"-D_GLIBCXX_USE_CXX11_ABI={abi}",
        "-DUSE_ROCM",
        f"-I{os.path.join(torch_dir, 'include')}",
        f"-I{os.path.join(torch_dir, 'include', 'torch', 'csrc', 'api', 'include')}",
        f"-I{sysconfig.get_paths()['include']}",
        f"-I{HIP_DIR}",
        "-x", "hip",
    ] + sources + [
        f"-L{os.path.join(torch_dir, 'lib')}",
        "-ltorch", "-ltorch_cpu", "-ltorch_hip", "-lc10", "-lc10_hip",
        "-ltorch_python", "-lamdhip64",
        f"-Wl,-rpath,{os.path.join(torch_dir, 'lib')}",
        "-o", OUT_SO,
    ]
    if verbose:
        print("[drla build]", " ".join(cmd))
    subprocess.run(cmd, check=True)
    # import self-check in a FRESH interpreter: a kernel missing from SOURCES
    # only surfaces as an undefined __device_stub__ symbol at import time
    repo_root = os.path.dirname(os.path.dirname(HERE))
    subprocess.run(
        [sys.executable, "-c",
         "import distributed_reinforcement_learning_amd.ops as o; "
         "assert o.available(), o._IMPORT_ERROR"],
        check=True, cwd=repo_root)
    if verbose:
        print(f"[drla build] wrote {OUT_SO} (import check OK)")
    return OUT_SO


if __name__ == "__main__":
    build(force="--force" in sys.argv)
