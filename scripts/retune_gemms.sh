#!/bin/bash
# Re-tune TunableOp on the current GEMM set (run on a GPU box), then merge
# the result back into profiles/ (copied via gpurun_out).
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME=/root/repo/gpurun_out/tunableop_gfx950_%d.csv
# seed with the existing table so old entries survive
cp profiles/tunableop_gfx950_0.csv gpurun_out/tunableop_gfx950_0.csv
export PYTORCH_TUNABLEOP_VERBOSE=0
timeout 900 python bench.py --steps 60 --warmup 40 > gpurun_out/retune_bench.log 2>&1
echo "tune rc=$?"
timeout 300 python bench.py --model resnet --steps 20 --warmup 10 >> gpurun_out/retune_bench.log 2>&1
echo "resnet tune rc=$?"
timeout 300 python bench.py --algo apex --steps 40 --warmup 10 >> gpurun_out/retune_bench.log 2>&1
echo "apex tune rc=$?"
timeout 400 python bench.py --algo r2d2 --steps 20 --warmup 5 >> gpurun_out/retune_bench.log 2>&1
echo "r2d2-80 tune rc=$?"
timeout 300 python bench.py --algo r2d2 --seq-len 15 --burn-in 7 --steps 40 --warmup 10 >> gpurun_out/retune_bench.log 2>&1
echo "r2d2-15 tune rc=$?"
wc -l gpurun_out/tunableop_gfx950_0.csv
