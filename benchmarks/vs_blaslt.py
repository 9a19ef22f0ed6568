#!/usr/bin/env python3
"""Hand-written g9 8-phase GEMM vs torch.matmul (hipBLASLt/rocBLAS) at the
training NT shapes. Evidence for whether the in-house kernel should carry
the plain large GEMMs (it fuses bias and feeds the split-K slab wgrads;
a library call would add back the epilogue round-trips)."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baton_amd.ops._ext import require_hip

ops = require_hip()


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


torch.manual_seed(0)
for name, M, N, K in [("sq4k", 4096, 4096, 4096), ("sq8k", 8192, 8192, 8192),
                      ("llama-q16", 8192, 4096, 4096),
                      ("gate16", 16384, 4096, 4096),
                      ("bert-ffn", 65536, 3072, 768)]:
    A = torch.randn(M, K, device="cuda").bfloat16().contiguous()
    B = torch.randn(N, K, device="cuda").bfloat16().contiguous()
    flops = 2.0 * M * N * K
    t_ours = bench(lambda: ops.gemm(A, B, 0))
    t_torch = bench(lambda: A @ B.t())
    print(f"{name:9s} [{M:5d}x{N:5d}x{K:5d}] ours {flops/t_ours/1e12:7.0f} TF"
          f"  torch/blaslt {flops/t_torch/1e12:7.0f} TF"
          f"  ratio {t_torch/t_ours:5.2f}x")
