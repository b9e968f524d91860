"""hipGraph-captured IMPALA train step.

The eager step is launch-bound on MI355X: ~300 kernel dispatches per step,
~1.9 ms GPU-busy inside a ~4 ms wall step (rocprof r01,
profiles/impala_bench_r01_kernels.md). Capturing the
normalize -> unroll -> V-trace -> backward -> fused-optimizer pipeline into
hipGraphs cuts the replay to ~1.3 ms GPU (measured, profiles/ triage r03).

Two measured host-side traps shape this class (gpu_triage.py r03/r04):
  * a mid-flight ``.item()`` on the loss tensors costs ~11 ms (blocking-sync
    wake latency), so losses are NOT read per step — ``last_losses()`` reads
    them on demand (logging cadence), and the steady-state loop never syncs;
  * staging numpy -> pinned is a ~4 ms single-thread memcpy, so callers that
    can produce data straight into ``self.pinned`` (the trajectory queue, the
    bench pool) skip it entirely.

Step structure (three graphs + a copy stream):
  host:     write decayed LR into a device buffer (never blocks)
  graph 1:  normalize frames, batched unroll (bf16), fused V-trace losses
            (forward only)
  copy str: the NEXT batch's pinned -> static-input H2D, ordered after
            graph 1 via an event, so the ~18 MB upload hides behind the
            backward. Every read of a static input happens INSIDE graph 1:
            the conv-l1 kernel stashes its u8 input in passing and the
            other backward-read inputs (previous_action / action /
            initial_c) are cloned at the top of the forward
  graph 2:  backward (autograd ASSIGNS scatter-mode grads — no
            AccumulateGrad adds)
  graph 3:  one-kernel grad gather + (world > 1: CAPTURED RCCL all-reduce)
            + fused global-norm clip + RMSProp update reading the LR buffer
  host:     record consume-event; losses stay on-device until
            last_losses()

Weights and optimizer state are snapshotted before the warmup iterations and
restored before capture, so graphing never perturbs training state.
"""

from __future__ import annotations

from typing import Dict, Optional, Tuple

import numpy as np
import torch


class GraphedImpalaStep:
    def __init__(self, agent, batch_size: int, warmup_iters: int = 3):
        assert agent.device.type == "cuda", "graphed step needs a GPU"
        # u8 conv layers stash their input in-kernel from here on: the
        # overlapped H2D rewrites the static input buffers during the
        # backward, and the l1 wgrad re-reads the input (see _fwd)
        from distributed_reinforcement_learning_amd.ops import conv_op
        conv_op.STASH_INPUTS = True
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
            # int32: the fused V-trace kernel consumes i32 and the ring
            # stores i32 — an i64 static input cost one [B,T] cast per
            # replay (actions feed nothing else in the captured step)
            "action": z((B, T), torch.int32),
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
        self._consumed = torch.cuda.Event()
        self._consumed.record()
        self._uploaded = torch.cuda.Event()
        self._uploaded.record()
        self._fwd_done = torch.cuda.Event()
        self._copy_stream = torch.cuda.Stream()
        self._primed = False

        # ---- warmup (eager, side stream), with state snapshot/restore -----
        opt = agent.optimizer
        # scatter-grad mode: backward ASSIGNS grads (no per-param
        # AccumulateGrad adds); one gather kernel packs them (optim.py)
        opt.enable_scatter_grads()
        snap_params = opt.flat_params.detach().clone()
        snap_state = {k: v.detach().clone()
                      for k, v in opt._state_tensors().items()}
        side = torch.cuda.Stream()
        side.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(side):
            for _ in range(warmup_iters):
                self._fwd_bwd()
                opt.gather_grads_eager()
                opt.step_tensor_lr(self.lr_buf)
        torch.cuda.current_stream().wait_stream(side)
        torch.cuda.synchronize()
        with torch.no_grad():
            opt.flat_params.copy_(snap_params)
            for k, v in opt._state_tensors().items():
                v.copy_(snap_state[k])
            opt.flat_grads.zero_()

        # ---- capture ------------------------------------------------------
        # forward and backward are SEPARATE graphs so the next step's H2D
        # upload (pinned -> static inputs, ~110 us) can overlap the
        # backward on a copy stream; _fwd() confines every input read to
        # graph 1 (conv-l1 input stash + clones of the backward-read
        # fields), so the overlap cannot tear the backward's reads.
        self.g_fwd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_fwd):
            self._total, self.losses = self._fwd()
        # preallocated backward seed: backward() without an explicit
        # gradient fills a scalar ones(()) INSIDE the graph (one fill
        # kernel per replay)
        self._bwd_seed = torch.ones((), device=dev)
        self.g_bwd = torch.cuda.CUDAGraph()
        with torch.cuda.graph(self.g_bwd, pool=self.g_fwd.pool()):
            # retain_graph: the saved tensors live in the shared capture
            # pool and are rewritten by every g_fwd replay
            self._total.backward(gradient=self._bwd_seed,
                                 retain_graph=True)
            # join the side-stream conv wgrads (ops/conv_op.py) so the
            # capture region ends with a single ordered stream
            from distributed_reinforcement_learning_amd.ops.conv_op import (
                join_wgrad_stream,
            )
            join_wgrad_stream()
        # .grad now holds capture-pool tensors at replay-stable addresses
        opt.build_gather_table()
        from distributed_reinforcement_learning_amd.parallel.dist import (
            is_distributed, world_size,
        )
        import os as _os
        self._distributed = is_distributed() and (
            world_size() > 1 or bool(_os.environ.get("DRLA_FORCE_DIST_GRAPH")))
        self._eager_reduce = False
        self.g_opt = torch.cuda.CUDAGraph()
        if self._distributed:
            # capture gather + RCCL all-reduce + update as ONE graph
            # (NCCL/RCCL collectives are capture-legal and every rank
            # replays in lockstep). Capture failure is FATAL by default —
            # a silent per-rank eager fallback deadlocks the lockstep
            # replay (parallel/dist.py handle_capture_failure).
            try:
                # prime the communicator outside capture
                agent.reduce_gradients()
                torch.cuda.synchronize()
                with torch.cuda.graph(self.g_opt, pool=self.g_fwd.pool()):
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
                with torch.cuda.graph(self.g_opt, pool=self.g_fwd.pool()):
                    opt.step_tensor_lr(self.lr_buf)
        else:
            with torch.cuda.graph(self.g_opt, pool=self.g_fwd.pool()):
                # single-GPU: the gather rides inside the optimizer graph
                opt.gather_grads()
                opt.step_tensor_lr(self.lr_buf)

    def _fwd(self):
        agent = self.agent
        i = dict(self.inputs)
        # The overlapped H2D upload (copy stream) is ordered only after
        # the FORWARD, but four inputs are re-read by backward kernels
        # (conv-l1 wgrad: state; embed scatter: previous_action; V-trace
        # bwd: action; LSTM tail bwd: initial_c) — clone those INSIDE the
        # forward graph so every read of a static buffer happens before
        # the upload barrier. ~19 MB through the capture pool, ~6 us.
        # (state is NOT cloned: the conv-l1 forward kernel bundles a
        # pass-through stash of its input instead — ops/conv_op.py
        # STASH_INPUTS, ~4 us bundled vs ~20 us standalone clone)
        import os as _os
        if _os.environ.get("DRLA_NO_INPUT_CLONE") != "1":  # measurement-only escape
            for k in ("previous_action", "action", "initial_c"):
                i[k] = i[k].clone()
        s = agent.prepare_frames(i["state"])
        pi_loss, baseline_loss, entropy, total = agent.compute_losses(
            s, i["reward"], i["action"], i["done"], i["behavior_policy"],
            i["previous_action"], i["initial_h"], i["initial_c"])
        return total, (pi_loss.detach(), baseline_loss.detach(),
                       entropy.detach())

    def _fwd_bwd(self) -> Tuple[torch.Tensor, ...]:
        agent = self.agent
        if not agent.optimizer.scatter:
            agent.optimizer.flat_grads.zero_()
        total, losses = self._fwd()
        total.backward()
        return losses

    # -- input staging -------------------------------------------------------

    def stage_to_pinned(self, batch: Dict[str, np.ndarray]) -> None:
        """Host-side memcpy of a numpy batch into the pinned buffers
        (~4 ms for the reference shape; skip by filling self.pinned
        directly, e.g. TrajectoryQueue.sample_batch_into)."""
        for k, pin in self.pinned.items():
            pin.copy_(torch.as_tensor(np.asarray(batch[k])))

    def upload_inputs(self, src: Optional[Dict[str, torch.Tensor]] = None
                      ) -> None:
        """Synchronous-path upload (priming / non-pipelined callers)."""
        self._consumed.synchronize()
        src = src or self.pinned
        for k, dst in self.inputs.items():
            dst.copy_(src[k], non_blocking=True)
        self._uploaded.record()
        self._primed = True

    def _upload_overlapped(self, src: Dict[str, torch.Tensor]) -> None:
        """Queue the NEXT step's H2D on the copy stream, ordered after this
        step's forward (the only consumer of the static inputs)."""
        with torch.cuda.stream(self._copy_stream):
            self._copy_stream.wait_event(self._fwd_done)
            for k, dst in self.inputs.items():
                dst.copy_(src[k], non_blocking=True)
            self._uploaded.record(self._copy_stream)

    def wait_pinned_free(self) -> None:
        """Block until the last async H2D has finished reading the pinned
        staging — call before overwriting self.pinned."""
        self._uploaded.synchronize()

    # -- stepping ------------------------------------------------------------

    def step(self, batch: Optional[Dict[str, np.ndarray]] = None,
             pinned_src: Optional[Dict[str, torch.Tensor]] = None
             ) -> Tuple[torch.Tensor, ...]:
        """Run one full train step. Sources, in priority order: ``batch``
        (numpy, staged through self.pinned), ``pinned_src`` (caller-owned
        pinned tensors), or self.pinned already filled. Never syncs; returns
        the loss TENSORS (device). Read them with last_losses() at logging
        cadence.

        Pipelined upload (pinned paths only): the FIRST call uploads
        synchronously; later calls replay on the inputs uploaded during the
        PREVIOUS step's backward and queue this call's data on the copy
        stream — one-batch latency, which the asynchronous actor-learner
        loop does not notice (losses/scalars lag one batch). The numpy
        ``batch=`` path stays strictly synchronous."""
        agent = self.agent
        main = torch.cuda.current_stream()
        if batch is not None:
            # numpy path keeps same-batch semantics (no pipelining) — the
            # eager-parity test and train()-style callers rely on it
            self.stage_to_pinned(batch)
            self.upload_inputs()
            src = None
        else:
            src = pinned_src if pinned_src is not None else self.pinned
            if not self._primed:
                self.upload_inputs(src)
                src = None
        main.wait_event(self._uploaded)
        lr = agent.lr_at(agent.global_step)
        self.lr_buf.fill_(lr)
        self.g_fwd.replay()
        self._fwd_done.record(main)
        if src is not None:
            # overlap this call's H2D with the backward; the replay above
            # consumed the PREVIOUS call's upload (one-batch pipeline)
            self._upload_overlapped(src)
        self.g_bwd.replay()
        if self._distributed and self._eager_reduce:
            agent.optimizer.gather_grads()
            agent.reduce_gradients()
        self.g_opt.replay()
        self._consumed.record()
        agent.optimizer.step_count += 1
        agent.global_step += 1
        agent.num_env_frames += int(np.prod(self.inputs["reward"].shape))
        self._last_lr = lr
        return self.losses

    def last_losses(self) -> Tuple[float, float, float, float]:
        """(pi_loss, baseline_loss, entropy, lr) of the most recent completed
        step — syncs the device."""
        pi, bl, ent = (float(x) for x in self.losses)
        return pi, bl, ent, getattr(self, "_last_lr", 0.0)
