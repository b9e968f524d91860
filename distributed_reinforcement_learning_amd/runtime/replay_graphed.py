"""hipGraph-captured PER learner step (Ape-X / R2D2).

One graph replays sample -> gather -> loss forward+backward -> priority
update; a second replays grad-gather (+ captured RCCL all-reduce in DP) +
the fused Adam update. Everything that changes across steps is read from
device memory at replay time: the segment tree and payloads (mutated by the
eager ingest between steps), beta / n_entries (device buffers the host
advances), the RNG state (philox offsets are graph-managed), and lr_t (a
1-element buffer).

Same warmup snapshot/restore discipline as runtime/graphed.py: weights,
optimizer state, the priority tree and beta are restored after the eager
warmup iterations so graph construction never perturbs training."""

from __future__ import annotations

import os
from typing import Callable, Dict, Tuple

import torch


class GraphedReplayStep:
    def __init__(self, agent, memory, batch_size: int,
                 loss_fn: Callable[[Dict[str, torch.Tensor], torch.Tensor],
                                   Tuple[torch.Tensor, torch.Tensor]],
                 warmup_iters: int = 3):
        assert agent.device.type == "cuda"
        self.agent = agent
        self.memory = memory
        self.batch_size = batch_size
        self._loss_fn = loss_fn
        dev = agent.device
        self.lr_buf = torch.zeros(1, dtype=torch.float32, device=dev)

        opt = agent.optimizer
        opt.enable_scatter_grads()
        snap_params = opt.flat_params.detach().clone()
        snap_state = {k: v.detach().clone()
                      for k, v in opt._state_tensors().items()}
        snap_tree = memory.tree.detach().clone()
        snap_beta = memory.beta

        def _iter():
            rows, idxs, w = memory.sample_static(batch_size)
            b = memory.gather(rows)
            loss, td = loss_fn(b, w)
            loss.backward()
            from distributed_reinforcement_learning_amd.ops.conv_op import (
                join_wgrad_stream,
            )
            join_wgrad_stream()  # side-stream conv wgrads (capture-safe)
            memory.update_batch(idxs, td)
            return loss.detach(), td

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                _iter()
                opt.gather_grads_eager()
                opt.step_tensor_lr(self.lr_buf)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        with torch.no_grad():
            opt.flat_params.copy_(snap_params)
            for k, v in opt._state_tensors().items():
                v.copy_(snap_state[k])
            opt.flat_grads.zero_()
            memory.tree.copy_(snap_tree)
        memory.beta = snap_beta
        memory.beta_buf.fill_(snap_beta)

        self.g_main = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_main):
            self.loss, self.td = _iter()
        opt.build_gather_table()

        from distributed_reinforcement_learning_amd.parallel.dist import (
            is_distributed, world_size,
        )
        self._distributed = is_distributed() and (
            world_size() > 1
            or bool(os.environ.get("DRLA_FORCE_DIST_GRAPH")))
        self._eager_reduce = False
        self.g_opt = torch.cuda.CUDAGraph()
        if self._distributed:
            try:
                agent.reduce_gradients()
                torch.cuda.synchronize()
                with torch.cuda.graph(self.g_opt, pool=self.g_main.pool()):
                    opt.gather_grads()
                    agent.reduce_gradients()
                    opt.step_tensor_lr(self.lr_buf)
            except Exception as exc:
                from distributed_reinforcement_learning_amd.parallel.dist import (
                    handle_capture_failure,
                )
                handle_capture_failure(exc)
                self._eager_reduce = True
                self.g_opt = torch.cuda.CUDAGraph()
                with torch.cuda.graph(self.g_opt, pool=self.g_main.pool()):
                    opt.step_tensor_lr(self.lr_buf)
        else:
            with torch.cuda.graph(self.g_opt, pool=self.g_main.pool()):
                opt.gather_grads()
                opt.step_tensor_lr(self.lr_buf)

    REBUILD_EVERY = 2048  # steps between interior-sum drift repairs

    def step(self) -> torch.Tensor:
        """One sample+train+update step; never syncs. Returns the loss
        TENSOR (float() it only at logging cadence)."""
        agent = self.agent
        opt = agent.optimizer
        self._steps = getattr(self, "_steps", 0) + 1
        if self._steps % self.REBUILD_EVERY == 0:
            # eager, between replays: repair float32 tree drift
            self.memory.rebuild()
        lr = agent.lr_at(agent.global_step)
        opt.step_count += 1
        self.lr_buf.fill_(opt.lr_t_for(lr, opt.step_count)
                          if hasattr(opt, "lr_t_for") else lr)
        self.memory.advance_beta()
        self.g_main.replay()
        if self._distributed and self._eager_reduce:
            opt.gather_grads()
            agent.reduce_gradients()
        self.g_opt.replay()
        agent.global_step += 1
        return self.loss


class GraphedTdScore:
    """hipGraph-captured ingest-side TD scoring (Ape-X get_td_error / R2D2
    get_td_error_batch): arriving unrolls have a fixed shape, and the
    no-grad scoring forward is ~half the learner loop's eager launches.
    ``score_fn(inputs) -> td tensor`` must be pure no-grad compute over the
    static input buffers."""

    def __init__(self, agent, example: Dict[str, torch.Tensor],
                 score_fn: Callable[[Dict[str, torch.Tensor]],
                                    torch.Tensor]):
        assert agent.device.type == "cuda"
        self.inputs = {k: torch.zeros_like(v) for k, v in example.items()}
        self._score_fn = score_fn
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(2):
                score_fn(self.inputs)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        self.graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.graph):
            self.td = score_fn(self.inputs)

    def score(self, batch: Dict[str, torch.Tensor]) -> torch.Tensor:
        """Copies ``batch`` into the static buffers, replays, and returns
        the TD tensor (plus self.inputs holds the staged batch — feed BOTH
        straight into GpuMemory.add_batch)."""
        for k, dst in self.inputs.items():
            dst.copy_(batch[k], non_blocking=True)
        self.graph.replay()
        return self.td


class GraphedTrainStep:
    """hipGraph-captured plain learner step (A3C flavor): static input
    buffers -> loss forward+backward as one graph, grad-gather (+ captured
    RCCL all-reduce in DP) + fused optimizer as a second. Same scatter-grad
    and snapshot/restore discipline as GraphedReplayStep."""

    def __init__(self, agent, example: Dict[str, torch.Tensor],
                 loss_fn: Callable[[Dict[str, torch.Tensor]],
                                   Tuple[torch.Tensor, ...]],
                 warmup_iters: int = 3):
        assert agent.device.type == "cuda"
        self.agent = agent
        self._loss_fn = loss_fn
        dev = agent.device
        self.inputs = {k: v.detach().clone() for k, v in example.items()}
        self.lr_buf = torch.zeros(1, dtype=torch.float32, device=dev)

        opt = agent.optimizer
        opt.enable_scatter_grads()
        snap_params = opt.flat_params.detach().clone()
        snap_state = {k: v.detach().clone()
                      for k, v in opt._state_tensors().items()}

        def _iter():
            losses = loss_fn(self.inputs)
            losses[-1].backward()
            from distributed_reinforcement_learning_amd.ops.conv_op import (
                join_wgrad_stream,
            )
            join_wgrad_stream()  # side-stream conv wgrads (capture-safe)
            return tuple(x.detach() for x in losses[:-1])

        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                _iter()
                opt.gather_grads_eager()
                opt.step_tensor_lr(self.lr_buf)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        with torch.no_grad():
            opt.flat_params.copy_(snap_params)
            for k, v in opt._state_tensors().items():
                v.copy_(snap_state[k])
            opt.flat_grads.zero_()

        self.g_main = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_main):
            self.losses = _iter()
        opt.build_gather_table()

        from distributed_reinforcement_learning_amd.parallel.dist import (
            is_distributed, world_size,
        )
        self._distributed = is_distributed() and (
            world_size() > 1
            or bool(os.environ.get("DRLA_FORCE_DIST_GRAPH")))
        self._eager_reduce = False
        self.g_opt = torch.cuda.CUDAGraph()
        if self._distributed:
            try:
                agent.reduce_gradients()
                torch.cuda.synchronize()
                with torch.cuda.graph(self.g_opt, pool=self.g_main.pool()):
                    opt.gather_grads()
                    agent.reduce_gradients()
                    opt.step_tensor_lr(self.lr_buf)
            except Exception as exc:
                from distributed_reinforcement_learning_amd.parallel.dist import (
                    handle_capture_failure,
                )
                handle_capture_failure(exc)
                self._eager_reduce = True
                self.g_opt = torch.cuda.CUDAGraph()
                with torch.cuda.graph(self.g_opt, pool=self.g_main.pool()):
                    opt.step_tensor_lr(self.lr_buf)
        else:
            with torch.cuda.graph(self.g_opt, pool=self.g_main.pool()):
                opt.gather_grads()
                opt.step_tensor_lr(self.lr_buf)

    def step(self, batch: Dict[str, torch.Tensor]):
        """Copy batch into the static inputs and replay; returns the loss
        TENSORS plus this step's lr (floats only at logging cadence)."""
        agent = self.agent
        opt = agent.optimizer
        for k, dst in self.inputs.items():
            dst.copy_(batch[k], non_blocking=True)
        lr = agent.lr_at(agent.global_step)
        opt.step_count += 1
        self.lr_buf.fill_(opt.lr_t_for(lr, opt.step_count)
                          if hasattr(opt, "lr_t_for") else lr)
        self.g_main.replay()
        if self._distributed and self._eager_reduce:
            opt.gather_grads()
            agent.reduce_gradients()
        self.g_opt.replay()
        agent.global_step += 1
        return self.losses + (lr,)
