#!/usr/bin/env python3
"""A/B the g8 schedule variants (BATON_G8_SCHED env) on NT shapes."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baton_amd.ops._ext import require_hip

ops = require_hip()
S = os.environ.get("BATON_G8_SCHED", "0")


def bench(fn, reps=15, warm=5):
    for _ in range(warm):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / reps / 1000


for name, M, N, K in [("sq4k", 4096, 4096, 4096), ("sq8k", 8192, 8192, 8192),
                      ("llama-q16", 8192, 4096, 4096),
                      ("gate16", 8192, 14336, 4096)]:
    A = (torch.rand(M, K, device="cuda") * 2 - 1).bfloat16()
    B = (torch.rand(N, K, device="cuda") * 2 - 1).bfloat16()
    t = bench(lambda: ops.gemm(A, B, 0))
    print(f"SCHED={S} {name}: {2*M*N*K/t/1e12:.0f} TF", flush=True)
