"""hipGraph-captured IMPALA train step.

The eager step is launch-bound on MI355X: ~300 kernel dispatches per step,
~1.9 ms GPU-busy inside a ~4.6 ms wall step (rocprof r01,
profiles/impala_bench_r01_kernels.md). Capturing the whole
normalize -> unroll -> V-trace -> backward -> fused-optimizer pipeline into a
hipGraph collapses the host-side launch gaps to one graph replay.

Structure per step:
  host:    copy batch into pinned staging, async H2D into the graph's static
           input buffers, write the decayed LR into a 1-element device buffer
  graph 1: zero flat grads, normalize frames, batched unroll (bf16),
           V-trace, losses, backward (into the flat grad bucket)
  eager:   single fused RCCL all-reduce of the flat bucket (world > 1) —
           kept outside the graph so capture needs no collective support
  graph 2: global-norm clip + RMSProp update reading the LR buffer

Weights and optimizer state are snapshotted before the warmup iterations and
restored before capture, so graphing never perturbs training state.
"""

from __future__ import annotations

from typing import Dict, Tuple

import numpy as np
import torch


class GraphedImpalaStep:
    def __init__(self, agent, batch_size: int, warmup_iters: int = 3):
        assert agent.device.type == "cuda", "graphed step needs a GPU"
        self.agent = agent
        B, T = batch_size, agent.trajectory
        A, H = agent.num_action, agent.lstm_hidden_size
        HH, WW, C = agent.input_shape
        dev = agent.device

        def z(shape, dtype):
            return torch.zeros(shape, dtype=dtype, device=dev)

        self.inputs: Dict[str, torch.Tensor] = {
            "state": z((B, T, HH, WW, C), torch.uint8),
            "reward": z((B, T), torch.float32),
            "action": z((B, T), torch.int64),
            "done": z((B, T), torch.bool),
            "behavior_policy": z((B, T, A), torch.float32),
            "previous_action": z((B, T), torch.int64),
            "initial_h": z((B, T, H), torch.float32),
            "initial_c": z((B, T, H), torch.float32),
        }
        self.pinned: Dict[str, torch.Tensor] = {
            k: torch.empty_like(v, device="cpu").pin_memory()
            for k, v in self.inputs.items()
        }
        self.lr_buf = torch.zeros(1, dtype=torch.float32, device=dev)

        # ---- warmup (eager, side stream), with state snapshot/restore -----
        opt = agent.optimizer
        snap_params = opt.flat_params.detach().clone()
        snap_state = {k: v.detach().clone()
                      for k, v in opt._state_tensors().items()}
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self._fwd_bwd()
                opt.step_tensor_lr(self.lr_buf)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        with torch.no_grad():
            opt.flat_params.copy_(snap_params)
            for k, v in opt._state_tensors().items():
                v.copy_(snap_state[k])
            opt.flat_grads.zero_()

        # ---- capture ------------------------------------------------------
        self.g_fwd_bwd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_fwd_bwd):
            self.losses = self._fwd_bwd()
        self.g_opt = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_opt):
            opt.step_tensor_lr(self.lr_buf)

        from distributed_reinforcement_learning_amd.parallel.dist import (
            is_distributed,
        )
        self._distributed = is_distributed()

    def _fwd_bwd(self) -> Tuple[torch.Tensor, ...]:
        from distributed_reinforcement_learning_amd.ops import normalize_frames
        agent = self.agent
        agent.optimizer.flat_grads.zero_()
        i = self.inputs
        s = normalize_frames(i["state"])
        pi_loss, baseline_loss, entropy, total = agent.compute_losses(
            s, i["reward"], i["action"], i["done"], i["behavior_policy"],
            i["previous_action"], i["initial_h"], i["initial_c"])
        total.backward()
        return (pi_loss.detach(), baseline_loss.detach(), entropy.detach())

    def load_inputs(self, batch: Dict[str, np.ndarray]) -> None:
        """Stage a host batch into the graph's static input buffers."""
        for k, dst in self.inputs.items():
            src = batch[k]
            if isinstance(src, torch.Tensor) and src.is_cuda:
                dst.copy_(src, non_blocking=True)
            else:
                pin = self.pinned[k]
                pin.copy_(torch.as_tensor(np.asarray(src)))
                dst.copy_(pin, non_blocking=True)

    def step(self, batch: Dict[str, np.ndarray]) -> Tuple[float, ...]:
        """Run one full train step; returns (pi_loss, baseline_loss,
        entropy, lr) as floats (one sync at the end)."""
        agent = self.agent
        self.load_inputs(batch)
        lr = agent.lr_at(agent.global_step)
        self.lr_buf.fill_(lr)
        self.g_fwd_bwd.replay()
        if self._distributed:
            agent.reduce_gradients()
        self.g_opt.replay()
        agent.optimizer.step_count += 1
        agent.global_step += 1
        agent.num_env_frames += int(np.prod(self.inputs["reward"].shape))
        pi, bl, ent = (float(x) for x in self.losses)
        return pi, bl, ent, lr
