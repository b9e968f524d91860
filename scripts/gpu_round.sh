#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
timeout 900 python scripts/gpu_triage.py staged > gpurun_out/triage2.log 2>&1
echo "triage rc=$?"
cat gpurun_out/triage2.log
