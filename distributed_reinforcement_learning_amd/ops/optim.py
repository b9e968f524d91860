"""Fused optimizers with global-norm clipping (K12 in SURVEY.md §2.5).

Replaces the reference's tf.clip_by_global_norm + RMSProp/Adam apply
(agent/impala.py:96-100, agent/apex.py:73-77) with:

1. one fused squared-norm reduction over ALL grads (flat view), then
2. one fused update kernel applying clip-scale + RMSProp/Adam in a single
   HBM pass per parameter bucket.

Parameters and grads live in ONE contiguous flat buffer each (built once at
optimizer construction; module parameters are views into it), so the whole
update is two kernel launches — and the DP all-reduce (parallel/dist.py)
reduces the same flat grad buffer with a single RCCL call.

Mixed precision: when the model's parameters are bf16 (the learner's GPU
models), the optimizer keeps an fp32 MASTER copy; the fused kernel updates
the master and writes the rounded bf16 copy the model computes with — no
per-layer weight casts anywhere in the step, and the all-reduce moves bf16
(half the xGMI bytes).

CPU fallback implements identical math in torch (used by the unit tests as
the golden reference).
"""

from __future__ import annotations

from typing import Iterable, List, Optional

import torch

from distributed_reinforcement_learning_amd import ops as _ops


def _is_cl4(p: torch.Tensor) -> bool:
    """4D channels_last param (conv weights): flat-buffer slots store its
    PHYSICAL [CO,KH,KW,CI] order so the custom conv path's
    permute(0,2,3,1).reshape view is copy-free both ways (the strided
    permute copies were ~6 kernels = ~30 us/step, trace r02)."""
    return p.dim() == 4 and p.is_contiguous(
        memory_format=torch.channels_last) and not p.is_contiguous()


def _phys_flat(p: torch.Tensor) -> torch.Tensor:
    """Flat view of p in its physical memory order."""
    if _is_cl4(p):
        return p.permute(0, 2, 3, 1).reshape(-1)
    return p.reshape(-1)


def flatten_dense_params(params: List[torch.Tensor]):
    """Re-home ``params`` as views into one new contiguous flat buffer.

    Slots are 8-element aligned (small zero holes between params) so the
    scatter-mode gather kernel can use vec8 loads/stores; returns
    (flat, slots) with slots = [(offset, numel), ...]. Channels_last 4D
    params keep their PHYSICAL layout in the slot (p stays a
    channels_last strided view into the flat buffer)."""
    slots = []
    offset = 0
    for p in params:
        slots.append((offset, p.numel()))
        offset += (p.numel() + 7) & ~7
    flat = torch.zeros(offset, dtype=params[0].dtype,
                       device=params[0].device)
    for p, (off, n) in zip(params, slots):
        flat[off:off + n].copy_(_phys_flat(p.detach()))
        if _is_cl4(p):
            co, ci, kh, kw = p.shape
            p.data = flat[off:off + n].view(co, kh, kw, ci) \
                .permute(0, 3, 1, 2)
        else:
            p.data = flat[off:off + n].view_as(p)
    return flat, slots


class _FlatOptimizerBase:
    """Holds flat param/grad/state buffers; subclasses implement the update.

    ``flat_params`` is the model's compute copy (fp32 or bf16);
    ``master`` is the fp32 copy the update math runs on (aliases
    flat_params when params are already fp32)."""

    def __init__(self, params: Iterable[torch.Tensor], lr: float,
                 clip_norm: Optional[float]):
        self.params = [p for p in params if p.requires_grad]
        assert len(self.params) > 0
        self.flat_params, self.slots = flatten_dense_params(self.params)
        self.flat_grads = torch.zeros_like(self.flat_params)
        self.mixed = self.flat_params.dtype != torch.float32
        self.master = (self.flat_params.detach().float()
                       if self.mixed else self.flat_params)
        self.scatter = False
        self._gather_table = None
        # route autograd into the flat grad buffer (channels_last params
        # get a matching strided view so accumulation lands in physical
        # slot order)
        for p, (off, n) in zip(self.params, self.slots):
            p.grad = self._slot_view(p, self.flat_grads, off, n)
        self.lr = lr
        self.clip_norm = clip_norm
        self.step_count = 0

    @staticmethod
    def _slot_view(p: torch.Tensor, flat: torch.Tensor, off: int,
                   n: int) -> torch.Tensor:
        if _is_cl4(p):
            co, ci, kh, kw = p.shape
            return flat[off:off + n].view(co, kh, kw, ci).permute(0, 3, 1, 2)
        return flat[off:off + n].view_as(p)

    def zero_grad(self) -> None:
        self.flat_grads.zero_()

    # -- scatter-grad mode (graphed GPU learner) ---------------------------
    # With .grad routed into flat views, every param costs one AccumulateGrad
    # add kernel per backward (~20 x 4.5 us/step at the small-kernel floor,
    # profile r23). Scatter mode leaves .grad = None so backward ASSIGNS the
    # grad tensors (no kernels); under hipGraph capture those tensors live at
    # stable pool addresses, and ONE gather kernel packs them into
    # flat_grads (which the DP all-reduce and the fused update consume).

    def enable_scatter_grads(self) -> None:
        self.scatter = True
        self._gather_table = None
        for p in self.params:
            p.grad = None

    def gather_grads_eager(self) -> None:
        """Eager/warmup-path gather: per-param copies, then detach again."""
        with torch.no_grad():
            for p, (off, n) in zip(self.params, self.slots):
                self.flat_grads[off:off + n].copy_(_phys_flat(p.grad))
                p.grad = None

    def build_gather_table(self) -> None:
        """Call once right after hipGraph capture of backward: .grad now
        holds capture-pool tensors whose addresses are replay-stable."""
        dev = self.flat_grads.device
        ptrs, offs, sizes = [], [], []
        for p, (off, n) in zip(self.params, self.slots):
            g = p.grad
            # channels_last grads are one contiguous memory block in the
            # slot's (physical) order — the linear gather copy is correct
            dense = g is not None and (g.is_contiguous() or _is_cl4(g))
            if not dense or g.numel() != n \
                    or g.dtype != self.flat_grads.dtype:
                raise RuntimeError(
                    f"scatter-grad table: bad grad for slot {off} "
                    f"(shape {None if g is None else tuple(g.shape)})")
            ptrs.append(g.data_ptr())
            offs.append(off)
            sizes.append(n)
        self._gather_table = (
            torch.tensor(ptrs, dtype=torch.int64, device=dev),
            torch.tensor(offs, dtype=torch.int64, device=dev),
            torch.tensor(sizes, dtype=torch.int64, device=dev))
        # persistent sq-norm partials (16 slots x 16-float cache lines),
        # zeroed by the gather kernel each step
        self._norm_ws = torch.zeros(256, dtype=torch.float32, device=dev)

    def gather_grads(self) -> None:
        """One-kernel scattered-grad -> flat_grads pack (also re-zeroes
        the persistent sq-norm buffer for the update that follows)."""
        ext = _ops.require_ext()
        srcs, offs, sizes = self._gather_table
        ext.grad_gather(srcs, offs, sizes, self.flat_grads, self._norm_ws)

    def grad_global_norm(self) -> torch.Tensor:
        if self.flat_grads.is_cuda:
            ext = _ops.require_ext()
            if self.flat_grads.dtype == torch.bfloat16:
                return ext.sq_norm_bf16(self.flat_grads).sum().sqrt()
            return ext.sq_norm(self.flat_grads).sum().sqrt()
        return self.flat_grads.float().norm()

    def _clip_scale(self) -> torch.Tensor:
        """tf.clip_by_global_norm semantics: scale = clip/max(norm, clip)."""
        norm = self.grad_global_norm()
        if self.clip_norm is None:
            return torch.ones_like(norm)
        return self.clip_norm / torch.clamp(norm, min=self.clip_norm)

    def _sync_model_from_master(self) -> None:
        if self.mixed:
            with torch.no_grad():
                self.flat_params.copy_(self.master)

    def refresh_master(self) -> None:
        """Re-derive the fp32 master from the (bf16) model copy — REQUIRED
        after any out-of-band rewrite of the parameters (rank-0 broadcast,
        checkpoint restore without optimizer state): the update math runs
        on the master, so a stale master would silently revert the rewrite
        on the next step and diverge DP ranks."""
        if self.mixed:
            with torch.no_grad():
                self.master.copy_(self.flat_params.float())

    def state_dict(self) -> dict:
        sd = {"step_count": self.step_count,
              "state": {k: v for k, v in self._state_tensors().items()}}
        if self.mixed:
            sd["master"] = self.master
        return sd

    def load_state_dict(self, sd: dict) -> None:
        self.step_count = sd["step_count"]
        for k, v in sd["state"].items():
            self._state_tensors()[k].copy_(v)
        if self.mixed and "master" in sd:
            self.master.copy_(sd["master"])
            self._sync_model_from_master()

    def _state_tensors(self) -> dict:
        raise NotImplementedError

    def step(self, lr: Optional[float] = None) -> None:
        raise NotImplementedError

    def step_tensor_lr(self, lr_buf: torch.Tensor) -> None:
        """GPU-only, hipGraph-capturable update: lr is read from a 1-element
        device buffer the host rewrites before each replay."""
        raise NotImplementedError


class FusedRMSProp(_FlatOptimizerBase):
    """TF-RMSProp semantics: ms = rho*ms + (1-rho)*g^2,
    p -= lr * g / sqrt(ms + eps) — eps inside the sqrt, as TF does.
    Matches reference agent/impala.py:97 (decay=.99, momentum=0, eps=.1)."""

    def __init__(self, params, lr: float, rho: float = 0.99,
                 eps: float = 0.1, clip_norm: Optional[float] = None):
        super().__init__(params, lr, clip_norm)
        self.rho = rho
        self.eps = eps
        self.ms = torch.zeros_like(self.master)

    def _state_tensors(self):
        return {"ms": self.ms}

    @torch.no_grad()
    def step(self, lr: Optional[float] = None) -> None:
        lr = self.lr if lr is None else lr
        if self.flat_params.is_cuda:
            lr_buf = torch.full((1,), lr, dtype=torch.float32,
                                device=self.flat_params.device)
            self.step_tensor_lr(lr_buf)
        else:
            g = self.flat_grads.float() * self._clip_scale()
            self.ms.mul_(self.rho).addcmul_(g, g, value=1 - self.rho)
            self.master.addcdiv_(g, (self.ms + self.eps).sqrt(), value=-lr)
            self._sync_model_from_master()
        self.step_count += 1

    def step_tensor_lr(self, lr_buf: torch.Tensor) -> None:
        ext = _ops.require_ext()
        clip = float(self.clip_norm or -1.0)
        if self.mixed:
            ext.rmsprop_step_bf16_t(self.flat_params, self.flat_grads,
                                    self.master, self.ms, clip, lr_buf,
                                    self.rho, self.eps,
                                    getattr(self, "_norm_ws", None))
        else:
            ext.rmsprop_step_t(self.flat_params, self.flat_grads, self.ms,
                               clip, lr_buf, self.rho, self.eps)


class FusedAdam(_FlatOptimizerBase):
    """TF-AdamOptimizer semantics (reference agent/apex.py:73): bias-corrected
    lr_t = lr * sqrt(1-b2^t)/(1-b1^t); p -= lr_t * m / (sqrt(v) + eps)."""

    def __init__(self, params, lr: float, beta1: float = 0.9,
                 beta2: float = 0.999, eps: float = 1e-8,
                 clip_norm: Optional[float] = None):
        super().__init__(params, lr, clip_norm)
        self.beta1, self.beta2, self.eps = beta1, beta2, eps
        self.m = torch.zeros_like(self.master)
        self.v = torch.zeros_like(self.master)

    def _state_tensors(self):
        return {"m": self.m, "v": self.v}

    def lr_t_for(self, lr: float, t: int) -> float:
        return lr * (1 - self.beta2 ** t) ** 0.5 / (1 - self.beta1 ** t)

    @torch.no_grad()
    def step(self, lr: Optional[float] = None) -> None:
        lr = self.lr if lr is None else lr
        self.step_count += 1
        lr_t = self.lr_t_for(lr, self.step_count)
        if self.flat_params.is_cuda:
            lr_buf = torch.full((1,), lr_t, dtype=torch.float32,
                                device=self.flat_params.device)
            self.step_tensor_lr(lr_buf)
        else:
            g = self.flat_grads.float() * self._clip_scale()
            self.m.mul_(self.beta1).add_(g, alpha=1 - self.beta1)
            self.v.mul_(self.beta2).addcmul_(g, g, value=1 - self.beta2)
            self.master.addcdiv_(self.m, self.v.sqrt() + self.eps,
                                 value=-lr_t)
            self._sync_model_from_master()

    def step_tensor_lr(self, lr_buf: torch.Tensor) -> None:
        """lr_buf must already hold the bias-corrected lr_t (lr_t_for)."""
        ext = _ops.require_ext()
        clip = float(self.clip_norm or -1.0)
        if self.mixed:
            ext.adam_step_bf16_t(self.flat_params, self.flat_grads,
                                 self.master, self.m, self.v, clip, lr_buf,
                                 self.beta1, self.beta2, self.eps)
        else:
            ext.adam_step_t(self.flat_params, self.flat_grads, self.m,
                            self.v, clip, lr_buf, self.beta1, self.beta2,
                            self.eps)