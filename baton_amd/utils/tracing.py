"""Tracing / profiling helpers.

The reference's observability is print() + tqdm. Here:

  * ``trace_scope(name)`` — roctx range (torch.cuda.nvtx maps to roctxRange
    on ROCm) around round phases, so `rocprofv3 --marker-trace` attributes
    kernel time to {local_train, fwd, bwd, opt, fedavg_reduce, broadcast}.
    Enabled when BATON_TRACE=1 (zero overhead otherwise).
  * ``PhaseTimer`` — wall-clock per named phase with running stats,
    exported by the manager /metrics endpoint and worker /status.
"""

from __future__ import annotations

import contextlib
import os
import time
from collections import defaultdict
from typing import Dict

import torch

_TRACE = os.environ.get("BATON_TRACE", "0") == "1"


def tracing_enabled() -> bool:
    return _TRACE


@contextlib.contextmanager
def trace_scope(name: str):
    if _TRACE and torch.cuda.is_available():
        torch.cuda.nvtx.range_push(name)   # roctxRangePush on ROCm
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield


class PhaseTimer:
    """Accumulates wall time per phase: with timer.phase('fedavg'): ..."""

    def __init__(self) -> None:
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)

    @contextlib.contextmanager
    def phase(self, name: str):
        t0 = time.perf_counter()
        try:
            yield
        finally:
            dt = time.perf_counter() - t0
            self.totals[name] += dt
            self.counts[name] += 1

    def summary(self) -> Dict[str, dict]:
        return {
            k: {
                "total_sec": self.totals[k],
                "count": self.counts[k],
                "mean_ms": 1000.0 * self.totals[k] / max(self.counts[k], 1),
            }
            for k in self.totals
        }

    def reset(self) -> None:
        self.totals.clear()
        self.counts.clear()
