#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/gpu_tests.log 2>&1
echo "gpu tests rc=$?"; tail -6 gpurun_out/gpu_tests.log
# end-to-end spawn-mode runs ON GPU: learner on cuda:0 with GPU replay,
# CPU actors, shm transport
timeout 1500 python -m pytest tests/test_end_to_end.py -q > gpurun_out/e2e_gpu.log 2>&1
echo "e2e-on-gpu rc=$?"; tail -4 gpurun_out/e2e_gpu.log
timeout 900 python bench.py --steps 150 --warmup 30 > gpurun_out/bench_graph.log 2>&1
echo "bench rc=$?"; cat gpurun_out/bench_graph.log
