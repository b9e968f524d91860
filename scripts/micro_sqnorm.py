import torch
import sys, os; sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from distributed_reinforcement_learning_amd import ops as _ops
ext = _ops.require_ext()
x = torch.randn(4_100_000, device="cuda").to(torch.bfloat16)
for _ in range(20):
    ext.sq_norm_bf16(x)
torch.cuda.synchronize()
s, e = torch.cuda.Event(True), torch.cuda.Event(True)
s.record()
for _ in range(200):
    out = ext.sq_norm_bf16(x)
e.record(); torch.cuda.synchronize()
print(f"sq_norm_bf16: {s.elapsed_time(e) / 200 * 1000:.1f} us")
print("parity:", torch.allclose(out.sum(), x.float().pow(2).sum(), rtol=2e-2))
