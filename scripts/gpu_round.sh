#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
export TMPDIR=/tmp
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
timeout 1500 python -m pytest tests/ -q -m gpu > gpurun_out/gpu_tests.log 2>&1
echo "gpu tests rc=$?"; tail -3 gpurun_out/gpu_tests.log
timeout 900 python -m pytest tests/test_end_to_end.py -q > gpurun_out/e2e.log 2>&1
echo "e2e rc=$?"; tail -2 gpurun_out/e2e.log
timeout 900 python bench.py --steps 300 --warmup 50 > gpurun_out/bench_graph.log 2>&1
echo "bench rc=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_graph.log; grep -o '"ms_per_step": [0-9.]*' gpurun_out/bench_graph.log
timeout 900 python bench.py --steps 60 --warmup 15 --model resnet > gpurun_out/bench_resnet.log 2>&1
echo "resnet rc=$?"; grep -o '"value": [0-9.]*' gpurun_out/bench_resnet.log
timeout 600 python -u scripts/bench_algos.py > gpurun_out/bench_algos.log 2>&1
echo "algos rc=$?"; cat gpurun_out/bench_algos.log | grep algo
timeout 600 python -m torch.distributed.run --nnodes=1 --nproc-per-node 1 --master-addr 127.0.0.1 --master-port 29571 bench.py --gpus 1 --steps 60 --warmup 15 > gpurun_out/bench_dist1.log 2>&1
echo "dist canary rc=$?"; grep -o '"ms_per_step": [0-9.]*' gpurun_out/bench_dist1.log
timeout 600 python -c "import __graft_entry__ as g; g.build(); g.smoke(); print('graft clean')" > gpurun_out/smoke.log 2>&1
echo "smoke rc=$?"; tail -2 gpurun_out/smoke.log
bash scripts/prof_bench.sh r35
