#!/usr/bin/env python3
"""A/B: routed TN wgrad (transpose both -> 8-phase split-K slabs) vs
native-TN split-K (stage_transposed staging, no transposes) at the BERT
dW shapes. Routed timings INCLUDE the two transpose kernels — that is
the end-to-end cost the model pays."""
import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baton_amd.ops._ext import require_hip

ops = require_hip()


def bench(fn, reps=20, warm=5):
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
    return t0.elapsed_time(t1) / reps


torch.manual_seed(0)
TOK = 65536
for name, M, N in [("qkv", 2304, 768), ("out", 768, 768),
                   ("ff1", 3072, 768), ("ff2", 768, 3072)]:
    dy = (torch.randn(TOK, M, device="cuda") * 0.1).bfloat16().contiguous()
    x = (torch.randn(TOK, N, device="cuda") * 0.1).bfloat16().contiguous()
    routed = bench(lambda: ops.gemm(dy, x, 2))
    native = bench(lambda: ops.gemm(dy, x, 2, None, False, False, 1.0, 0.0,
                                    None, True))
    a = ops.gemm(dy, x, 2, None, False, False, 1.0, 0.0, None, True)
    ref = dy.float().t() @ x.float()
    err = (a.float() - ref).abs().max().item()
    print(f"{name:4s} [{M:5d}x{N:5d}xK{TOK}] routed {routed*1e3:8.1f} us  "
          f"nativeTN {native*1e3:8.1f} us  ({routed/native:4.2f}x)  "
          f"maxerr {err:.3e}")
