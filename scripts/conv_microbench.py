#!/usr/bin/env python3
"""A/B microbench: custom MFMA conv kernels vs torch(MIOpen/CK) at the
flagship IMPALA learner shape (N = B*T = 640). Within-process interleaved
timing (guide §5.4 rule 24)."""

import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch
import torch.nn.functional as F

from distributed_reinforcement_learning_amd import ops

ROUNDS = 30
N = 640


def t_ms(fn, n=ROUNDS):
    for _ in range(5):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(n):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / n * 1e3


def main():
    ext = ops.require_ext()
    torch.manual_seed(0)
    results = []
    shapes = [
        (0, 4, 32, 8, 4, 84, 20, True),
        (2, 32, 64, 4, 2, 20, 9, False),
        (3, 64, 64, 3, 1, 9, 7, False),
    ]
    for layer, ci, co, k, s, hi, ho, u8 in shapes:
        conv = torch.nn.Conv2d(ci, co, k, stride=s).cuda()
        conv_bf = torch.nn.Conv2d(ci, co, k, stride=s).cuda()
        conv_bf.load_state_dict(conv.state_dict())
        conv_bf = conv_bf.to(torch.bfloat16).to(
            memory_format=torch.channels_last)
        conv_ref = conv.to(torch.bfloat16).to(
            memory_format=torch.channels_last)
        if u8:
            x = torch.randint(0, 256, (N, hi, hi, ci), dtype=torch.uint8,
                              device="cuda")
            x_ref = (x.float() / 255).bfloat16().permute(0, 3, 1, 2) \
                .contiguous(memory_format=torch.channels_last)
        else:
            x = (torch.randn(N, hi, hi, ci, device="cuda") * 0.5).bfloat16()
            x_ref = x.permute(0, 3, 1, 2).contiguous(
                memory_format=torch.channels_last)
        w_flat = conv_bf.weight.permute(0, 2, 3, 1).reshape(
            co, -1).contiguous()
        bias = conv_bf.bias.contiguous()
        xc = x.contiguous()

        t_custom = t_ms(lambda: ext.conv_fwd(layer, xc, w_flat, bias, False))
        t_torch = t_ms(lambda: F.relu(conv_ref(x_ref)))
        flops = 2.0 * N * ho * ho * co * k * k * ci
        results.append(
            f"fwd L{layer}: custom {t_custom*1e3:7.1f} us "
            f"({flops/t_custom/1e9:6.1f} TF) | torch {t_torch*1e3:7.1f} us "
            f"({flops/t_torch/1e9:6.1f} TF) | speedup "
            f"{t_torch/t_custom:4.2f}x")

        # backward A/B: wgrad+dgrad via our kernels vs autograd
        dy = torch.randn(N, ho, ho, co, device="cuda").bfloat16().contiguous()
        y, _ = ext.conv_fwd(layer, xc, w_flat, bias, False)
        def custom_bwd():
            dy_m = ext.relu_mask_bwd(dy, y, co, layer)
            ext.conv_wgrad(layer, xc, dy_m)
            if layer >= 2:
                ext.conv_dgrad(layer, dy_m, w_flat)
        x_ag = x_ref.detach().clone().requires_grad_(layer >= 2)
        conv_ag = conv_ref
        def torch_bwd():
            conv_ag.weight.grad = None
            conv_ag.bias.grad = None
            out = F.relu(conv_ag(x_ag))
            out.backward(dy.permute(0, 3, 1, 2).contiguous(
                memory_format=torch.channels_last))
        t_cb = t_ms(custom_bwd, n=15)
        t_tb = t_ms(torch_bwd, n=15)
        results.append(
            f"bwd L{layer}: custom {t_cb*1e3:7.1f} us | torch(autograd) "
            f"{t_tb*1e3:7.1f} us | speedup {t_tb/t_cb:4.2f}x")
    print("\n".join(results), flush=True)


if __name__ == "__main__":
    main()
