"""Op-level attribution of the eager IMPALA step via torch.profiler.

Graph replays are anonymous; this runs the SAME losses+backward+optimizer
eagerly so every remaining at::native kernel maps to an aten op. Prints the
top ops by device time.
"""
import sys, os
sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
import numpy as np
import torch

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

def one_step():
    agent.optimizer.zero_grad()
    s = agent.prepare_frames(batch["state"])
    pi, bl, ent, total = agent.compute_losses(
        s, batch["reward"], batch["action"], batch["done"],
        batch["behavior_policy"], batch["previous_action"],
        batch["initial_h"], batch["initial_c"])
    total.backward()
    agent.optimizer.step(6e-4)

for _ in range(5):
    one_step()
torch.cuda.synchronize()

from torch.profiler import profile, ProfilerActivity
with profile(activities=[ProfilerActivity.CPU, ProfilerActivity.CUDA]) as prof:
    for _ in range(10):
        one_step()
    torch.cuda.synchronize()

print(prof.key_averages().table(sort_by="self_cuda_time_total", row_limit=45,
                                max_name_column_width=60))
