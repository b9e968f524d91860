#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
timeout 1200 python -m pytest tests/ -q -m gpu > gpurun_out/gpu_tests.log 2>&1
echo "gpu tests rc=$?"; tail -4 gpurun_out/gpu_tests.log
timeout 900 python bench.py --steps 100 --warmup 20 > gpurun_out/bench_graph.log 2>&1
echo "bench-graph rc=$?"; cat gpurun_out/bench_graph.log
timeout 600 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('graft path clean')" > gpurun_out/smoke_func.log 2>&1
echo "smoke rc=$?"; tail -2 gpurun_out/smoke_func.log
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 900 rocprofv3 --kernel-trace --stats -d gpurun_out/prof3 -o graphed \
  -- python bench.py --steps 30 --warmup 10 > gpurun_out/rocprof3.log 2>&1
echo "rocprof rc=$?"
