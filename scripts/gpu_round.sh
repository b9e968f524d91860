#!/bin/bash
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950
timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"
timeout 900 python scripts/conv_microbench.py > gpurun_out/conv_microbench.log 2>&1
echo "microbench rc=$?"; cat gpurun_out/conv_microbench.log
cd /tmp && export TMPDIR=/tmp && cd /root/repo
rocprofv3 --list-avail > gpurun_out/counters.txt 2>&1 || true
grep -iE "MFMA|VALU_BUSY|LDS_BANK|FETCH_SIZE|WRITE_SIZE|GUI_ACTIVE" gpurun_out/counters.txt | head -30
timeout 900 rocprofv3 --kernel-trace --stats \
  --pmc SQ_INSTS_MFMA SQ_VALU_MFMA_BUSY_CYCLES SQ_BUSY_CYCLES SQ_INSTS_VALU SQ_LDS_BANK_CONFLICT \
  -d gpurun_out/pmc -o pmcrun -- python scripts/conv_microbench.py > gpurun_out/pmc.log 2>&1
echo "pmc rc=$?"; tail -3 gpurun_out/pmc.log; ls gpurun_out/pmc 2>/dev/null
