"""RCCL-over-xGMI data plane: the FedAvg theta-traffic fast path.

The reference moves every round's parameters twice through the manager's
single TCP socket as pickled CPU tensors (SURVEY.md §2.3/§5 — fine for
Linear(10,1), catastrophic for ResNet/BERT). Here the 8 GPUs of one MI355X
node are 8 federated clients, one process per GPU (`torch.distributed`,
backend "nccl" = RCCL on ROCm), and the theta traffic is:

    pre-scaled REDUCE to the root GPU  +  BROADCAST of the new global model

which matches the FedAvg semantics exactly (Sum n_i * theta_i on root, scale
by 1/N — reference manager.py:119-126) and suits the xGMI topology: links
are point-to-point (7 x ~153 GB/s per GPU), so the naturally-rooted
reduce/broadcast pair is used instead of translating an all-reduce pattern
designed for switched fabrics. Collectives run on a side stream so the next
round's first local batches can overlap the aggregation (overlap_stream).

Numerics: the reduce runs in fp32 regardless of model dtype (config
reduce_dtype), so the result is comparable against the CPU/HTTP FedAvg
oracle; tests assert near-bit-exact agreement in fp32.

CPU testing: the same class runs on gloo with world_size>1 (reduce/broadcast
are identical calls), which is how tests/test_data_plane.py exercises the
distributed path without a GPU.
"""

from __future__ import annotations

import logging
import os
from typing import List, Optional, Sequence

import torch
import torch.distributed as dist

from baton_amd.runtime.arena import FlatParamArena
from baton_amd.utils.config import DataPlaneConfig
from baton_amd.utils.tracing import trace_scope

log = logging.getLogger("baton.dataplane")


class FederatedDataPlane:
    """One instance per rank (= per GPU-client). Rank 0 is the aggregation
    root — the "manager GPU"."""

    def __init__(
        self,
        config: Optional[DataPlaneConfig] = None,
        device: Optional[torch.device] = None,
    ):
        self.config = config or DataPlaneConfig()
        if not dist.is_initialized():
            self._init_from_env()
        self.rank = dist.get_rank()
        self.world_size = dist.get_world_size()
        if device is not None:
            self.device = device
        elif torch.cuda.is_available():
            self.device = torch.device("cuda", int(os.environ.get("LOCAL_RANK", self.rank % max(torch.cuda.device_count(), 1))))
        else:
            self.device = torch.device("cpu")
        self._side_stream: Optional[torch.cuda.Stream] = None
        if self.device.type == "cuda" and self.config.overlap_stream:
            self._side_stream = torch.cuda.Stream(device=self.device)
        self._reduce_bufs: dict = {}

    def _init_from_env(self) -> None:
        backend = self.config.backend
        if backend == "nccl" and not torch.cuda.is_available():
            backend = "gloo"
        os.environ.setdefault("MASTER_ADDR", self.config.master_addr)
        os.environ.setdefault("MASTER_PORT", str(self.config.master_port))
        dist.init_process_group(backend=backend)
        log.info(
            "data plane up: backend=%s rank=%d/%d",
            backend,
            dist.get_rank(),
            dist.get_world_size(),
        )

    # -- collectives -----------------------------------------------------------

    def gather_weights(self, n_samples: int) -> torch.Tensor:
        """All-gather every client's sample count -> fp64 [world] tensor
        (tiny; the FedAvg weights)."""
        t = torch.tensor([float(n_samples)], dtype=torch.float64, device=self._coll_device())
        out = [torch.zeros_like(t) for _ in range(self.world_size)]
        dist.all_gather(out, t)
        return torch.cat(out).cpu()

    def _coll_device(self) -> torch.device:
        # gloo wants CPU tensors; nccl/RCCL wants device tensors
        if dist.get_backend() == "gloo":
            return torch.device("cpu")
        return self.device

    def fedavg_flat(
        self, flat: torch.Tensor, n_samples: int, weights: Optional[torch.Tensor] = None
    ) -> torch.Tensor:
        """In-place federated average of a flat tensor across all ranks.

        Every rank ends up with the new global value (reduce to root 0 in
        fp32, scale by 1/N on root, broadcast). Returns the gathered weight
        vector for reuse (loss-history weighting, heaviest-rank policy).
        """
        if weights is None:
            weights = self.gather_weights(n_samples)
        total = float(weights.sum().item())
        if total <= 0:
            raise ValueError("total sample weight must be positive")
        scale = float(n_samples) / total

        coll_dev = self._coll_device()
        key = (flat.numel(), str(coll_dev))
        if key not in self._reduce_bufs:
            self._reduce_bufs[key] = torch.empty(
                flat.numel(), dtype=torch.float32, device=coll_dev
            )
        buf = self._reduce_bufs[key]
        # pre-scale: buf = (n_i / N) * theta_i — fused HIP scale_cast on GPU
        src = flat.reshape(-1)
        assert src.is_contiguous(), "fedavg_flat needs a contiguous flat tensor" 
        if buf.device.type == "cuda" and src.device == buf.device:
            from baton_amd.ops._ext import require_hip

            require_hip().scale_cast(buf, src, scale)
        else:
            buf.copy_(src.to(buf.device, torch.float32))
            buf.mul_(scale)
        with trace_scope("fedavg_reduce"):
            dist.reduce(buf, dst=0, op=dist.ReduceOp.SUM)
        with trace_scope("fedavg_broadcast"):
            dist.broadcast(buf, src=0)
        if buf.device.type == "cuda" and src.device == buf.device:
            from baton_amd.ops._ext import require_hip

            require_hip().cast_copy(src, buf)
        else:
            src.copy_(buf.to(flat.device, flat.dtype))
        return weights

    def fedavg_arena(
        self, arena: FlatParamArena, n_samples: int
    ) -> torch.Tensor:
        """FedAvg the whole model held in a FlatParamArena: params + float
        buffers averaged (one collective per dtype group); integer buffers
        broadcast from the heaviest client (same policy as
        fed.aggregate.fedavg_)."""
        weights = self.gather_weights(n_samples)
        for g in arena.all_groups:
            self.fedavg_flat(g.flat, n_samples, weights=weights)
        # integer buffers: copy from the heaviest rank (deterministic
        # tie-break: lowest rank wins, matching fed.aggregate)
        int_bufs = [
            b for _, b in arena.model.named_buffers() if not b.is_floating_point()
        ]
        if int_bufs:
            heaviest = int(torch.argmax(weights).item())
            for b in int_bufs:
                t = b.detach()
                if dist.get_backend() == "gloo" and t.device.type != "cpu":
                    cpu = t.cpu()
                    dist.broadcast(cpu, src=heaviest)
                    t.copy_(cpu)
                else:
                    dist.broadcast(t, src=heaviest)
        return weights

    def fedavg_model(self, model: torch.nn.Module, n_samples: int) -> torch.Tensor:
        """Arena-free path: average params + float buffers tensor-by-tensor
        (used for models that cannot be re-parented)."""
        weights = self.gather_weights(n_samples)
        named = [
            t
            for t in list(model.parameters()) + list(model.buffers())
            if t.is_floating_point()
        ]
        flat = torch.cat([t.detach().reshape(-1).float() for t in named])
        self.fedavg_flat(flat, n_samples, weights=weights)
        off = 0
        with torch.no_grad():
            for t in named:
                n = t.numel()
                t.copy_(flat[off : off + n].view(t.shape).to(t.dtype))
                off += n
        int_bufs = [b for b in model.buffers() if not b.is_floating_point()]
        if int_bufs:
            heaviest = int(torch.argmax(weights).item())
            for b in int_bufs:
                t = b.detach()
                if dist.get_backend() == "gloo" and t.device.type != "cpu":
                    cpu = t.cpu()
                    dist.broadcast(cpu, src=heaviest)
                    t.copy_(cpu)
                else:
                    dist.broadcast(t, src=heaviest)
        return weights

    # -- loss bookkeeping -------------------------------------------------------

    def weighted_mean_losses(
        self, loss_history: Sequence[float], weights: torch.Tensor
    ) -> List[float]:
        """Sample-weighted mean of per-epoch losses across ranks (the
        manager-side loss history of the HTTP path, manager.py:127-130)."""
        h = torch.tensor(list(loss_history), dtype=torch.float64, device=self._coll_device())
        w = float(weights[self.rank].item())
        total = float(weights.sum().item())
        h.mul_(w / total)
        dist.all_reduce(h, op=dist.ReduceOp.SUM)
        return [float(x) for x in h.cpu()]

    def barrier(self) -> None:
        dist.barrier()

    @staticmethod
    def shutdown() -> None:
        if dist.is_initialized():
            dist.destroy_process_group()
