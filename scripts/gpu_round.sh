#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/gpu_tests.log 2>&1
echo "gpu tests rc=$?"; tail -4 gpurun_out/gpu_tests.log
timeout 1500 python -m pytest tests/test_end_to_end.py -q > gpurun_out/e2e_gpu.log 2>&1
echo "e2e-on-gpu rc=$?"; tail -3 gpurun_out/e2e_gpu.log
timeout 900 python bench.py --steps 150 --warmup 30 > gpurun_out/bench_graph.log 2>&1
echo "bench rc=$?"; cat gpurun_out/bench_graph.log
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/proflatest -o latest \
  -- python bench.py --steps 30 --warmup 10 > gpurun_out/rocproflatest.log 2>&1
echo "rocprof rc=$?"
timeout 900 python bench.py --steps 60 --warmup 15 --model resnet > gpurun_out/bench_resnet.log 2>&1
echo "bench-resnet rc=$?"; cat gpurun_out/bench_resnet.log
