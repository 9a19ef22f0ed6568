"""Fused optimizers.

The reference's optimizer is stock ``torch.optim.SGD`` driven per-batch
(/root/reference/demo.py:34,47). Here the step is a hand-written gfx950 HIP
kernel (csrc/optim.hip): one launch per contiguous buffer, fp32 math, with
momentum / weight-decay / Adam variants — and when the model's parameters
live in a FlatParamArena (runtime/arena.py) the whole step is ONE kernel
over one flat buffer, sized for HBM3E streaming (vectorized float4 access).

CPU fallback uses torch._foreach so the control-plane tests run GPU-free;
on a GPU box the HIP kernel is mandatory (ops/_ext.require_hip).
"""

from __future__ import annotations

import math
from typing import Iterable, List, Optional

import torch

from baton_amd.ops._ext import require_hip
from baton_amd.utils.config import TrainConfig


def _collect(params: Iterable[torch.nn.Parameter]) -> List[torch.nn.Parameter]:
    out = [p for p in params if p.requires_grad]
    if not out:
        raise ValueError("optimizer got an empty parameter list")
    return out


class _FusedOptimizerBase:
    """Steps either a list of nn.Parameters (one kernel per tensor) or a
    FlatParamArena (ONE kernel per dtype group — the production path)."""

    def __init__(self, params=None, arena=None):
        self.arena = arena
        if arena is not None:
            # (flat_params, flat_grads) per dtype group
            self.params = [g.flat for g in arena.groups]
            self._arena_grads = [g.grad for g in arena.groups]
            self.device = self.params[0].device
        else:
            self.params = _collect(params)
            self._arena_grads = None
            self.device = self.params[0].device
        self._use_hip = self.device.type == "cuda"
        # Arena + HIP: the step kernel writes zeros back to the flat grad
        # after consuming it (zero_grad fused), so zero_grad() is a no-op
        # (arena grads start zeroed; backward accumulates into them).
        self._fused_zero = self._use_hip and arena is not None
        if self._use_hip:
            require_hip()  # fail loudly up front, not at step N

    def zero_grad(self, set_to_none: bool = True) -> None:
        if self.arena is not None:
            if not self._fused_zero:
                self.arena.zero_grads()
            return
        for p in self.params:
            if p.grad is not None:
                if set_to_none:
                    p.grad = None
                else:
                    p.grad.detach_().zero_()

    def _grads(self) -> List[Optional[torch.Tensor]]:
        if self._arena_grads is not None:
            return self._arena_grads
        return [p.grad for p in self.params]


class FusedSGD(_FusedOptimizerBase):
    """SGD with optional momentum + weight decay, one HIP kernel per buffer.

    Semantics match torch.optim.SGD (and therefore the reference demo):
        g = grad + wd * p
        m = mu * m + g          (momentum buffer, if mu > 0)
        p = p - lr * (m if mu>0 else g)
    """

    def __init__(
        self,
        params: Optional[Iterable[torch.nn.Parameter]] = None,
        lr: float = 1e-3,
        momentum: float = 0.0,
        weight_decay: float = 0.0,
        arena=None,
    ):
        super().__init__(params, arena)
        self.lr = lr
        self.momentum = momentum
        self.weight_decay = weight_decay
        self.momentum_bufs: List[Optional[torch.Tensor]] = [
            torch.zeros_like(p, dtype=torch.float32) if momentum > 0 else None
            for p in self.params
        ]

    @classmethod
    def from_arena(cls, arena, **kw):
        return cls(arena=arena, **kw)

    @torch.no_grad()
    def step(self) -> None:
        grads = self._grads()
        if self._use_hip:
            ops = require_hip()
            for p, g, m in zip(self.params, grads, self.momentum_bufs):
                if g is None:
                    continue
                ops.sgd_step(
                    p.data if isinstance(p, torch.nn.Parameter) else p,
                    g,
                    m if m is not None else torch.empty(0, device=self.device),
                    self.lr,
                    self.momentum,
                    self.weight_decay,
                    self._fused_zero,
                )
            return
        # CPU fallback (same math)
        for p, g, m in zip(self.params, grads, self.momentum_bufs):
            if g is None:
                continue
            gf = g.float()
            if self.weight_decay:
                gf = gf.add(p.data.float(), alpha=self.weight_decay)
            if self.momentum > 0:
                m.mul_(self.momentum).add_(gf)
                gf = m
            p.data.add_(gf.to(p.dtype), alpha=-self.lr)


class FusedAdam(_FusedOptimizerBase):
    """Adam (bias-corrected), one HIP kernel per buffer; fp32 moments.

    Semantics match torch.optim.Adam:
        m = b1*m + (1-b1)*g ; v = b2*v + (1-b2)*g^2
        p -= lr * (m / (1-b1^t)) / (sqrt(v / (1-b2^t)) + eps)
    with decoupled-free weight decay (L2: g += wd * p), like torch's Adam.
    """

    def __init__(
        self,
        params: Optional[Iterable[torch.nn.Parameter]] = None,
        lr: float = 1e-3,
        betas=(0.9, 0.999),
        eps: float = 1e-8,
        weight_decay: float = 0.0,
        arena=None,
    ):
        super().__init__(params, arena)
        self.lr = lr
        self.beta1, self.beta2 = betas
        self.eps = eps
        self.weight_decay = weight_decay
        self.step_count = 0
        self.exp_avg = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]
        self.exp_avg_sq = [torch.zeros_like(p, dtype=torch.float32) for p in self.params]

    @classmethod
    def from_arena(cls, arena, **kw):
        return cls(arena=arena, **kw)

    @torch.no_grad()
    def step(self) -> None:
        self.step_count += 1
        bc1 = 1.0 - self.beta1**self.step_count
        bc2 = 1.0 - self.beta2**self.step_count
        grads = self._grads()
        if self._use_hip:
            ops = require_hip()
            for p, g, m, v in zip(self.params, grads, self.exp_avg, self.exp_avg_sq):
                if g is None:
                    continue
                ops.adam_step(
                    p.data if isinstance(p, torch.nn.Parameter) else p,
                    g, m, v,
                    self.lr, self.beta1, self.beta2, self.eps,
                    self.weight_decay, bc1, bc2, self._fused_zero,
                )
            return
        for p, g, m, v in zip(self.params, grads, self.exp_avg, self.exp_avg_sq):
            if g is None:
                continue
            gf = g.float()
            if self.weight_decay:
                gf = gf.add(p.data.float(), alpha=self.weight_decay)
            m.mul_(self.beta1).add_(gf, alpha=1 - self.beta1)
            v.mul_(self.beta2).addcmul_(gf, gf, value=1 - self.beta2)
            denom = (v / bc2).sqrt_().add_(self.eps)
            p.data.add_((m / bc1 / denom).to(p.dtype), alpha=-self.lr)


def make_optimizer(params: Iterable[torch.nn.Parameter], cfg: TrainConfig):
    if cfg.optimizer == "sgd":
        return FusedSGD(
            params, lr=cfg.lr, momentum=cfg.momentum, weight_decay=cfg.weight_decay
        )
    if cfg.optimizer == "adam":
        return FusedAdam(
            params,
            lr=cfg.lr,
            betas=tuple(cfg.adam_betas),
            eps=cfg.adam_eps,
            weight_decay=cfg.weight_decay,
        )
    raise ValueError(f"unknown optimizer {cfg.optimizer!r}")
