"""Sweep DRLA_WGRAD_SPLIT values per conv layer via subprocess microbench."""
import os, subprocess, sys

CODE = r'''
import os, sys, torch
sys.path.insert(0, "/root/repo")
from distributed_reinforcement_learning_amd import ops as _ops
ext = _ops.require_ext()
torch.manual_seed(0)
shapes = {0: ((640,84,84,4), (640,20,20,32)),
          2: ((640,20,20,32), (640,9,9,64)),
          3: ((640,9,9,64), (640,7,7,64))}
for layer, (ishape, oshape) in shapes.items():
    if layer == 0:
        x = torch.randint(0, 255, ishape, dtype=torch.uint8, device="cuda")
    else:
        x = torch.randn(ishape, device="cuda").to(torch.bfloat16)
    dy = torch.randn(oshape, device="cuda").to(torch.bfloat16)
    for _ in range(10):
        ext.conv_wgrad(layer, x, dy)
    torch.cuda.synchronize()
    s, e = torch.cuda.Event(True), torch.cuda.Event(True)
    s.record()
    for _ in range(100):
        ext.conv_wgrad(layer, x, dy)
    e.record(); torch.cuda.synchronize()
    print(f"L{layer}: {s.elapsed_time(e)/100*1000:.1f} us")
'''
for split in ["256", "512", "768", "1024", "1536"]:
    env = dict(os.environ, DRLA_WGRAD_SPLIT=split)
    r = subprocess.run(["python", "-c", CODE], env=env, capture_output=True,
                       text=True, timeout=300)
    print(f"split={split}:", " | ".join(r.stdout.split("\n")[:-1]) or r.stderr[-200:])
