"""Flat parameter arena — the MI355X-first memory layout.

With 288 GB of HBM3E per GPU the right layout is a handful of large
contiguous buffers, not thousands of small allocations: every parameter
(and every float buffer) of a model becomes a view into ONE flat tensor, and
gradients become views into a matching flat gradient tensor. That makes

  * the FedAvg pre-scaled reduce ONE RCCL collective on one contiguous
    buffer (no gather/scatter, no bucketing) — sized for the 7x ~153 GB/s
    point-to-point xGMI links;
  * the fused optimizer step ONE HIP kernel launch over one buffer
    (ops/csrc/optim.hip streams it at HBM rate);
  * the global-model broadcast ONE collective.

The reference has no equivalent (its "flat layout" is a pickled dict per
client); this is the from-scratch redesign of that traffic.
"""

from __future__ import annotations

from typing import Dict, Iterable, List, Optional, Tuple

import torch


class FlatParamArena:
    """Re-parents a model's parameters (and optionally float buffers) into
    flat contiguous storage.

    After construction:
      * ``arena.flat_params`` — 1-D tensor; every ``model.parameters()``
        tensor is a view into it (same dtype/device);
      * ``arena.flat_grads`` — matching 1-D tensor; ``p.grad`` views are
        pre-assigned so autograd accumulates directly into it;
      * ``arena.flat_buffers`` — 1-D fp32-or-original-dtype concat of the
        model's floating-point buffers (BatchNorm running stats), also
        re-parented, or None when the model has none / include_buffers=False.

    Integer buffers (e.g. ``num_batches_tracked``) stay where they are —
    they are copied, not averaged, by the aggregation policy.
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
        dtypes = {p.dtype for _, p in named}
        if len(dtypes) != 1:
            raise ValueError(f"arena requires a single param dtype, got {dtypes}")
        self.param_dtype = dtypes.pop()
        device = named[0][1].device

        total = sum(p.numel() for _, p in named)
        self.flat_params = torch.empty(total, dtype=self.param_dtype, device=device)
        self.flat_grads = (
            torch.zeros(total, dtype=self.param_dtype, device=device) if grads else None
        )
        self.param_slices: Dict[str, Tuple[int, int]] = {}

        offset = 0
        for name, p in named:
            n = p.numel()
            self.flat_params[offset : offset + n].copy_(p.detach().reshape(-1))
            p.data = self.flat_params[offset : offset + n].view(p.shape)
            if grads:
                p.grad = self.flat_grads[offset : offset + n].view(p.shape)
            self.param_slices[name] = (offset, offset + n)
            offset += n

        # Float buffers (running stats): their own flat tensor per dtype
        # group is overkill — models here keep them fp32.
        self.flat_buffers: Optional[torch.Tensor] = None
        self.buffer_slices: Dict[str, Tuple[int, int]] = {}
        if include_buffers:
            fbufs = [
                (n, b)
                for n, b in model.named_buffers()
                if b.is_floating_point()
            ]
            if fbufs:
                bdtypes = {b.dtype for _, b in fbufs}
                if len(bdtypes) != 1:
                    raise ValueError(f"mixed buffer dtypes {bdtypes}")
                btotal = sum(b.numel() for _, b in fbufs)
                self.flat_buffers = torch.empty(
                    btotal, dtype=bdtypes.pop(), device=device
                )
                off = 0
                # re-parent via module attribute so state_dict sees the view
                mod_map = dict(model.named_modules())
                for name, b in fbufs:
                    n = b.numel()
                    self.flat_buffers[off : off + n].copy_(b.detach().reshape(-1))
                    mod_name, _, attr = name.rpartition(".")
                    setattr(mod_map[mod_name], attr,
                            self.flat_buffers[off : off + n].view(b.shape))
                    self.buffer_slices[name] = (off, off + n)
                    off += n

    @property
    def numel(self) -> int:
        return self.flat_params.numel()

    def zero_grads(self) -> None:
        if self.flat_grads is not None:
            self.flat_grads.zero_()

    def load_flat(self, flat: torch.Tensor) -> None:
        """Install new global weights from a flat tensor (one copy)."""
        self.flat_params.copy_(flat.to(self.flat_params.dtype))

    def check_views(self) -> bool:
        """True iff every parameter still aliases the arena (a torch op that
        re-assigns .data would break the invariant)."""
        for name, p in self.model.named_parameters():
            if not p.requires_grad:
                continue
            lo, hi = self.param_slices[name]
            if p.data.data_ptr() != self.flat_params[lo:hi].data_ptr():
                return False
        return True
