"""setup.py — `python setup.py build_ext --inplace` compiles the in-tree
gfx950 HIP extension (ops/_drla_hip.so) via hipcc (no hipify, no JIT cache)."""

import os
import sys

from setuptools import setup, find_packages
from setuptools.command.build_ext import build_ext as _build_ext

sys.path.insert(0, os.path.dirname(os.path.abspath(__file__)))


class BuildHip(_build_ext):
    def run(self):
        os.environ.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
        from distributed_reinforcement_learning_amd.ops.build import build
        build(verbose=True)


setup(
    name="distributed_reinforcement_learning_amd",
    version="0.1.0",
    packages=find_packages(include=[
        "distributed_reinforcement_learning_amd",
        "distributed_reinforcement_learning_amd.*",
    ]),
    cmdclass={"build_ext": BuildHip},
    # dummy ext module entry so `build_ext` runs
    ext_modules=[],
)
