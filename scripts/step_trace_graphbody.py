"""Stack-level attribution of the GRAPHED IMPALA step's torch-glue kernels.

Graph replays are anonymous in profilers, but the captured body is exactly:
forward (prepare+unroll+fused V-trace) -> backward in scatter-grad mode ->
gather + fused RMSProp. This runs that same sequence eagerly (scatter mode
ON, grads ASSIGNED, fused optimizer with tensor LR) and profiles with
python stacks, so every remaining at::native fill/copy/elementwise kernel
maps to the source line that emits it (527 zero-fills +
621 elementwise + 516 copyBuffer per 103 steps are still ~15% of step
time).
"""
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

from distributed_reinforcement_learning_amd.utils import tunableop
tunableop.enable()

from distributed_reinforcement_learning_amd.agents import impala

B, T, A, H = 32, 20, 18, 256
agent = impala.Agent(
    trajectory=T, input_shape=[84, 84, 4], num_action=A,
    lstm_hidden_size=H, discount_factor=0.99, start_learning_rate=6e-4,
    end_learning_rate=0.0, learning_frame=10 ** 9, baseline_loss_coef=1.0,
    entropy_coef=0.05, gradient_clip_norm=40.0, reward_clipping="abs_one",
    device="cuda:0", seed=7)
rng = np.random.default_rng(3)
dev = agent.device
batch = dict(
    state=torch.as_tensor(rng.integers(0, 255, (B, T, 84, 84, 4),
                                       dtype=np.uint8)).to(dev),
    reward=torch.as_tensor(rng.normal(size=(B, T)).astype(np.float32)).to(dev),
    action=torch.as_tensor(rng.integers(0, A, (B, T))).to(dev).long(),
    done=torch.as_tensor(rng.random((B, T)) < 0.02).to(dev),
    behavior_policy=torch.full((B, T, A), 1.0 / A).to(dev),
    previous_action=torch.as_tensor(rng.integers(0, A, (B, T))).to(dev).long(),
    initial_h=torch.zeros(B, T, H).to(dev),
    initial_c=torch.zeros(B, T, H).to(dev),
)

opt = agent.optimizer
opt.enable_scatter_grads()
lr_buf = torch.full((1,), 6e-4, device=dev)


def fwd_bwd():
    s = agent.prepare_frames(batch["state"])
    pi, bl, ent, total = agent.compute_losses(
        s, batch["reward"], batch["action"], batch["done"],
        batch["behavior_policy"], batch["previous_action"],
        batch["initial_h"], batch["initial_c"])
    total.backward()


def one_step():
    # the captured sequence (runtime/graphed.py g_fwd + g_bwd + g_opt),
    # including the REAL one-kernel grad gather (not the eager per-slot
    # copies)
    fwd_bwd()
    opt.gather_grads()
    opt.step_tensor_lr(lr_buf)
    # keep backward ASSIGNING (no AccumulateGrad adds), as under capture;
    # stale table ptrs are fine — only kernel timings matter here
    for p in opt.params:
        p.grad = None


for _ in range(5):
    fwd_bwd()
    opt.gather_grads_eager()
    opt.step_tensor_lr(lr_buf)
torch.cuda.synchronize()
# grads now live at stable eager-cached addresses; the table path works
# outside capture too as long as autograd reuses them (same shapes/graph)
fwd_bwd()
opt.build_gather_table()
opt.gather_grads()
opt.step_tensor_lr(lr_buf)
torch.cuda.synchronize()

from torch.profiler import profile, ProfilerActivity

with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA],
             with_stack=True, record_shapes=True) as prof:
    for _ in range(10):
        one_step()
    torch.cuda.synchronize()

print("======== by op (device time) ========")
print(prof.key_averages().table(sort_by="self_cuda_time_total",
                                row_limit=45, max_name_column_width=60))
print("======== glue ops by shape ========")
ka = prof.key_averages(group_by_input_shape=True)
glue = [e for e in ka
        if e.key.split("::")[-1] in ("fill_", "zero_", "copy_", "add_",
                                     "mul_", "mul", "add", "cat", "sum",
                                     "to", "_to_copy", "clone",
                                     "contiguous", "div", "div_", "neg",
                                     "stack")
        and e.self_device_time_total > 0]
glue.sort(key=lambda e: -e.self_device_time_total)
for e in glue[:40]:
    print(f"--- {e.key}  self_cuda={e.self_device_time_total/1000:.3f}ms "
          f"calls={e.count} shapes={e.input_shapes}")

prof.export_stacks("gpurun_out/step_stacks_cuda.txt",
                   "self_cuda_time_total")
print("stacks exported")
