#!/bin/bash
# GPU round 3: graph overhead triage + smoke-crash bisect.
set -x
cd /root/repo
mkdir -p gpurun_out
export PYTORCH_ROCM_ARCH=gfx950

timeout 900 python -m distributed_reinforcement_learning_amd.ops.build > gpurun_out/build.log 2>&1
echo "build rc=$?"

timeout 900 python scripts/gpu_triage.py all > gpurun_out/triage.log 2>&1
echo "triage rc=$?"
cat gpurun_out/triage.log

# smoke teardown bisect
timeout 300 python -c "import torch; torch.zeros(4, device='cuda'); print('A ok')" > gpurun_out/bisect.log 2>&1
echo "A(torch-only) rc=$?" | tee -a gpurun_out/bisect.log
timeout 300 python -c "
import torch
from distributed_reinforcement_learning_amd.agents import impala
a = impala.Agent(trajectory=8, input_shape=[84,84,4], num_action=18,
    lstm_hidden_size=64, discount_factor=0.99, start_learning_rate=6e-4,
    end_learning_rate=0.0, learning_frame=10**9, baseline_loss_coef=1.0,
    entropy_coef=0.05, gradient_clip_norm=40.0, reward_clipping='abs_one',
    device='cuda:0', seed=0)
print('B ok (agent built)')" >> gpurun_out/bisect.log 2>&1
echo "B(agent-build) rc=$?" | tee -a gpurun_out/bisect.log
timeout 300 python -c "
import numpy as np, torch
from distributed_reinforcement_learning_amd.agents import impala
B,T,A,H = 4,8,18,64
a = impala.Agent(trajectory=T, input_shape=[84,84,4], num_action=A,
    lstm_hidden_size=H, discount_factor=0.99, start_learning_rate=6e-4,
    end_learning_rate=0.0, learning_frame=10**9, baseline_loss_coef=1.0,
    entropy_coef=0.05, gradient_clip_norm=40.0, reward_clipping='abs_one',
    device='cuda:0', seed=0)
rng = np.random.default_rng(0)
out = a.train(state=rng.integers(0,255,(B,T,84,84,4),dtype=np.uint8),
    reward=rng.normal(size=(B,T)).astype(np.float32),
    action=rng.integers(0,A,(B,T)).astype(np.int32),
    done=np.zeros((B,T),bool),
    behavior_policy=np.full((B,T,A),1/A,np.float32),
    previous_action=rng.integers(0,A,(B,T)).astype(np.int32),
    initial_h=np.zeros((B,T,H),np.float32), initial_c=np.zeros((B,T,H),np.float32))
print('C ok (train step)', out[0])" >> gpurun_out/bisect.log 2>&1
echo "C(train-step) rc=$?" | tee -a gpurun_out/bisect.log
timeout 300 python -c "import __graft_entry__ as g; g.smoke(); print('D ok')" >> gpurun_out/bisect.log 2>&1
echo "D(graft-smoke) rc=$?" | tee -a gpurun_out/bisect.log
tail -20 gpurun_out/bisect.log
