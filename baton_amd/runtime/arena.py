"""Flat parameter arena — the MI355X-first memory layout.

With 288 GB of HBM3E per GPU the right layout is a handful of large
contiguous buffers, not thousands of small allocations: parameters (and
float buffers) become views into one flat tensor PER DTYPE GROUP (bf16
weights and fp32 BatchNorm params stay separate), and gradients become
views into matching flat gradient tensors. That makes

  * the FedAvg pre-scaled reduce ONE RCCL collective per dtype group on a
    contiguous buffer (no gather/scatter, no bucketing) — sized for the
    7 x ~153 GB/s point-to-point xGMI links;
  * the fused optimizer step ONE HIP kernel launch per group
    (ops/csrc/optim.hip streams it at HBM rate);
  * the global-model broadcast one collective per group.

The reference has no equivalent (its "flat layout" is a pickled dict per
client); this is the from-scratch redesign of that traffic.
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional, Tuple

import torch


@dataclass
class ArenaGroup:
    dtype: torch.dtype
    flat: torch.Tensor                      # parameters, 1-D
    grad: Optional[torch.Tensor]            # matching flat gradient, 1-D
    slices: Dict[str, Tuple[int, int]] = field(default_factory=dict)


class FlatParamArena:
    """Re-parents a model's parameters (and optionally float buffers) into
    flat contiguous storage, one group per dtype.

    After construction:
      * ``arena.groups`` — list of ArenaGroup; every trainable parameter is
        a view into its group's ``flat``, its ``.grad`` a view into
        ``grad`` (autograd accumulates straight into the flat buffer);
      * ``arena.buffer_groups`` — same for floating-point buffers
        (BatchNorm running stats), no grads.

    Integer buffers (``num_batches_tracked``) stay untouched — the
    aggregation policy copies, never averages, them.
    """

    def __init__(
        self,
        model: torch.nn.Module,
        include_buffers: bool = True,
        grads: bool = True,
    ):
        self.model = model
        named = [(n, p) for n, p in model.named_parameters() if p.requires_grad]
        if not named:
            raise ValueError("model has no trainable parameters")

        by_dtype: Dict[torch.dtype, List[Tuple[str, torch.nn.Parameter]]] = {}
        for n, p in named:
            by_dtype.setdefault(p.dtype, []).append((n, p))

        self.groups: List[ArenaGroup] = []
        self.param_group_of: Dict[str, int] = {}
        for dtype, items in by_dtype.items():
            device = items[0][1].device
            total = sum(p.numel() for _, p in items)
            flat = torch.empty(total, dtype=dtype, device=device)
            gflat = torch.zeros(total, dtype=dtype, device=device) if grads else None
            group = ArenaGroup(dtype, flat, gflat)
            offset = 0
            for name, p in items:
                n = p.numel()
                flat[offset : offset + n].copy_(p.detach().reshape(-1))
                p.data = flat[offset : offset + n].view(p.shape)
                if grads:
                    p.grad = gflat[offset : offset + n].view(p.shape)
                group.slices[name] = (offset, offset + n)
                self.param_group_of[name] = len(self.groups)
                offset += n
            self.groups.append(group)

        # Floating-point buffers (running stats), grouped by dtype as well.
        self.buffer_groups: List[ArenaGroup] = []
        if include_buffers:
            fbufs = [
                (n, b) for n, b in model.named_buffers() if b.is_floating_point()
            ]
            by_dtype_b: Dict[torch.dtype, List[Tuple[str, torch.Tensor]]] = {}
            for n, b in fbufs:
                by_dtype_b.setdefault(b.dtype, []).append((n, b))
            mod_map = dict(model.named_modules())
            for dtype, items in by_dtype_b.items():
                device = items[0][1].device
                total = sum(b.numel() for _, b in items)
                flat = torch.empty(total, dtype=dtype, device=device)
                group = ArenaGroup(dtype, flat, None)
                off = 0
                for name, b in items:
                    n = b.numel()
                    flat[off : off + n].copy_(b.detach().reshape(-1))
                    mod_name, _, attr = name.rpartition(".")
                    setattr(mod_map[mod_name], attr, flat[off : off + n].view(b.shape))
                    group.slices[name] = (off, off + n)
                    off += n
                self.buffer_groups.append(group)

    # -- convenience -----------------------------------------------------------

    @property
    def numel(self) -> int:
        return sum(g.flat.numel() for g in self.groups)

    @property
    def all_groups(self) -> List[ArenaGroup]:
        return self.groups + self.buffer_groups

    def zero_grads(self) -> None:
        for g in self.groups:
            if g.grad is not None:
                g.grad.zero_()

    def check_views(self) -> bool:
        """True iff every parameter still aliases its arena group (a torch
        op that re-assigns .data would break the invariant)."""
        for name, p in self.model.named_parameters():
            if not p.requires_grad:
                continue
            g = self.groups[self.param_group_of[name]]
            lo, hi = g.slices[name]
            if p.data.data_ptr() != g.flat[lo:hi].data_ptr():
                return False
        return True
