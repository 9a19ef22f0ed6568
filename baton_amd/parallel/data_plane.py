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
designed for switched fabrics.

Overlap (overlap_stream=True, GPU+nccl only): the whole aggregation runs on
a side HIP stream, bucketed. Each bucket's pre-scale (HIP scale_cast) and
cast-back (cast_copy) execute on the side stream while the RCCL reduce /
broadcast of the *previous* bucket is still in flight on the communicator's
own stream (async_op collectives; xGMI link time hides under the fused
casts, and vice versa). The compute stream is fenced with HIP events: it
waits for aggregation-complete only at `PendingAggregation.wait()` — or at
the end of `fedavg_arena` when called synchronously — so the next round's
parameter-independent work (grad zero-fill, input staging, host-side
bookkeeping) proceeds while theta traffic is still on the wire.

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

# Bucket size for the overlapped aggregation pipeline (elements of fp32).
# 16 MiB buckets: large enough that each RCCL call is bandwidth-bound on a
# single xGMI link (~153 GB/s per direction), small enough that the fused
# scale/cast of bucket i+1 has real link time to hide under.
_BUCKET_ELEMS = 4 * 1024 * 1024


class PendingAggregation:
    """Handle for an in-flight (side-stream) federated aggregation.

    `wait()` fences the CALLER's current stream against aggregation
    completion (no host sync). Anything enqueued on the compute stream
    before `wait()` overlaps the collectives; the first op that reads the
    averaged parameters must come after it."""

    def __init__(self, weights: torch.Tensor, done_event=None):
        self.weights = weights
        self._done = done_event
        self._waited = done_event is None

    def wait(self) -> torch.Tensor:
        if not self._waited:
            torch.cuda.current_stream().wait_event(self._done)
            self._waited = True
        return self.weights


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
        # handle for the most recent aggregation (set by fedavg_arena)
        self.pending: Optional[PendingAggregation] = None
        # (graph, captured weights, captured n_samples) from capture_aggregation
        self._agg_graph = None

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

        src = flat.reshape(-1)
        assert src.is_contiguous(), "fedavg_flat needs a contiguous flat tensor"
        buf = self._reduce_buf(src.numel())
        if buf.device.type == "cuda" and src.device == buf.device:
            if self._side_stream is not None:
                # overlapped path: pipeline on the side stream, fence here
                ev = torch.cuda.Event()
                ev.record(torch.cuda.current_stream())
                with torch.cuda.stream(self._side_stream):
                    self._side_stream.wait_event(ev)
                    self._fedavg_gpu_pipelined(src, buf, scale)
                    done = torch.cuda.Event()
                    done.record(self._side_stream)
                torch.cuda.current_stream().wait_event(done)
            else:
                self._fedavg_gpu_pipelined(src, buf, scale)
        else:
            buf.copy_(src.to(buf.device, torch.float32))
            buf.mul_(scale)
            with trace_scope("fedavg_reduce"):
                dist.reduce(buf, dst=0, op=dist.ReduceOp.SUM)
            with trace_scope("fedavg_broadcast"):
                dist.broadcast(buf, src=0)
            src.copy_(buf.to(flat.device, flat.dtype))
        return weights

    def _reduce_buf(self, numel: int) -> torch.Tensor:
        coll_dev = self._coll_device()
        key = (numel, str(coll_dev))
        if key not in self._reduce_bufs:
            self._reduce_bufs[key] = torch.empty(
                numel, dtype=torch.float32, device=coll_dev
            )
        return self._reduce_bufs[key]

    def _fedavg_gpu_pipelined(self, src: torch.Tensor, buf: torch.Tensor,
                              scale: float) -> None:
        """Bucketed scale -> reduce -> broadcast -> cast-back, enqueued on
        the CALLER's current stream (the side stream in overlap mode).

        Overlap structure: the fused HIP scale_cast of bucket i+1 runs on
        this stream while RCCL moves bucket i on the communicator's own
        stream (async_op=True); the cast-back of bucket i overlaps the
        collectives of buckets > i. RCCL executes r0,b0,r1,b1,... in issue
        order, so each broadcast reads its completed reduce without any
        extra fencing (reference semantics: manager.py:119-126)."""
        from baton_amd.ops._ext import require_hip

        ops = require_hip()
        n = src.numel()
        bcast_works = []
        bounds = list(range(0, n, _BUCKET_ELEMS))
        with trace_scope("fedavg_reduce_bcast"):
            for off in bounds:
                end = min(off + _BUCKET_ELEMS, n)
                piece = buf[off:end]
                # pre-scale: buf = (n_i / N) * theta_i — fused HIP kernel
                ops.scale_cast(piece, src[off:end], scale)
                dist.reduce(piece, dst=0, op=dist.ReduceOp.SUM, async_op=True)
                bcast_works.append(
                    dist.broadcast(piece, src=0, async_op=True)
                )
            for off, w in zip(bounds, bcast_works):
                end = min(off + _BUCKET_ELEMS, n)
                w.wait()  # stream-level fence only (no host sync)
                ops.cast_copy(src[off:end], buf[off:end])

    # -- hipGraph-captured aggregation ----------------------------------------

    def capture_aggregation(self, arena: FlatParamArena, n_samples: int):
        """hipGraph-capture the WHOLE per-round aggregation sequence (the
        manager-side round loop on the data plane): per-group bucketed
        scale -> RCCL reduce -> broadcast -> cast-back plus the integer-
        buffer broadcasts, replayed as one graph per round.

        Valid while every rank's sample count stays fixed (the FedAvg
        scale is baked into the captured scale_cast); ``fedavg_arena``
        checks the weights each round and falls back to eager when they
        change. Requires GPU + a non-gloo backend."""
        if self.device.type != "cuda" or dist.get_backend() == "gloo":
            return False
        weights = self.gather_weights(n_samples)
        total = float(weights.sum().item())
        scale = float(n_samples) / total
        heaviest = int(torch.argmax(weights).item())
        int_bufs = [
            b for _, b in arena.model.named_buffers() if not b.is_floating_point()
        ]

        def _sequence():
            stream = self._side_stream or torch.cuda.current_stream()
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            with torch.cuda.stream(stream):
                stream.wait_event(ev)
                for g in arena.all_groups:
                    src = g.flat.reshape(-1)
                    self._fedavg_gpu_pipelined(
                        src, self._reduce_buf(src.numel()), scale
                    )
                for b in int_bufs:
                    dist.broadcast(b.detach(), src=heaviest)
                done = torch.cuda.Event()
                done.record(stream)
            torch.cuda.current_stream().wait_event(done)

        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                _sequence()
        torch.cuda.current_stream().wait_stream(s)
        graph = torch.cuda.CUDAGraph()
        with torch.cuda.graph(graph):
            _sequence()
        self._agg_graph = (graph, weights.clone(), float(n_samples))
        log.info("aggregation hipGraph captured (%d groups, %d int buffers)",
                 len(arena.all_groups), len(int_bufs))
        return True

    def fedavg_arena(
        self, arena: FlatParamArena, n_samples: int, async_handle: bool = False
    ) -> torch.Tensor:
        """FedAvg the whole model held in a FlatParamArena: params + float
        buffers averaged (one bucketed pipeline per dtype group); integer
        buffers broadcast from the heaviest client (same policy as
        fed.aggregate.fedavg_).

        async_handle=True (GPU overlap mode only) skips the final fence and
        returns a :class:`PendingAggregation` via ``self.pending``; the
        caller overlaps parameter-independent work with the collectives and
        calls ``pending.wait()`` before the first read of the averaged
        model. The returned value is always the weight vector."""
        weights = self.gather_weights(n_samples)
        total = float(weights.sum().item())
        if total <= 0:
            raise ValueError("total sample weight must be positive")
        if self._agg_graph is not None:
            graph, wcap, ncap = self._agg_graph
            if float(n_samples) == ncap and torch.equal(weights, wcap):
                graph.replay()
                # the captured sequence ends with the compute-stream fence
                self.pending = PendingAggregation(weights)
                return weights
            log.warning("sample weights changed — aggregation graph dropped")
            self._agg_graph = None
        scale = float(n_samples) / total
        heaviest = int(torch.argmax(weights).item())
        int_bufs = [
            b for _, b in arena.model.named_buffers() if not b.is_floating_point()
        ]
        self.pending = PendingAggregation(weights)  # pre-fenced default

        on_gpu = (
            self.device.type == "cuda"
            and dist.get_backend() != "gloo"
            and all(g.flat.device == self.device for g in arena.all_groups)
        )
        if on_gpu and self._side_stream is not None:
            ev = torch.cuda.Event()
            ev.record(torch.cuda.current_stream())
            with torch.cuda.stream(self._side_stream):
                self._side_stream.wait_event(ev)
                for g in arena.all_groups:
                    src = g.flat.reshape(-1)
                    self._fedavg_gpu_pipelined(
                        src, self._reduce_buf(src.numel()), scale
                    )
                for b in int_bufs:
                    dist.broadcast(b.detach(), src=heaviest)
                done = torch.cuda.Event()
                done.record(self._side_stream)
            self.pending = PendingAggregation(weights, done)
            if not async_handle:
                self.pending.wait()
            return weights

        for g in arena.all_groups:
            self.fedavg_flat(g.flat, n_samples, weights=weights)
        # integer buffers: copy from the heaviest rank (deterministic
        # tie-break: lowest rank wins, matching fed.aggregate)
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
        manager-side loss history of the HTTP path, manager.py:127-130).

        Histories may have unequal lengths across ranks (matching the HTTP
        path's weighted_loss_history): ranks agree on the max length first,
        zero-pad, and carry a per-epoch weight denominator so each epoch
        averages over exactly the ranks that reported it."""
        dev = self._coll_device()
        n = len(loss_history)
        ln = torch.tensor([float(n)], dtype=torch.float64, device=dev)
        dist.all_reduce(ln, op=dist.ReduceOp.MAX)
        length = int(ln.cpu().item())
        if length == 0:
            return []
        w = float(weights[self.rank].item())
        h = torch.zeros(2, length, dtype=torch.float64, device=dev)
        if n:
            h[0, :n] = torch.tensor(
                [float(x) for x in loss_history], dtype=torch.float64
            ).to(dev) * w
            h[1, :n] = w
        dist.all_reduce(h, op=dist.ReduceOp.SUM)
        num, den = h[0].cpu(), h[1].cpu()
        return [
            float(num[e] / den[e]) if float(den[e]) > 0 else float("nan")
            for e in range(length)
        ]

    def barrier(self) -> None:
        dist.barrier()

    @staticmethod
    def shutdown() -> None:
        if dist.is_initialized():
            dist.destroy_process_group()
