#!/usr/bin/env python
"""Microbenchmark the MFMA GEMM kernels (TF at training-relevant shapes).

    python benchmarks/gemm_bench.py          # on a GPU box

Within-run timing via hip events around K repetitions per shape; reports
TFLOP/s per layout. Random uniform operands (guide §5.4 rule 25: never
quote zero-filled numbers).
"""

import os
import sys

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from baton_amd.ops._ext import require_hip


def bench(fn, reps=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = torch.cuda.Event(enable_timing=True)
    t1 = torch.cuda.Event(enable_timing=True)
    t0.record()
    for _ in range(reps):
        fn()
    t1.record()
    torch.cuda.synchronize()
    return t0.elapsed_time(t1) / reps / 1000.0  # sec


def main():
    ops = require_hip()
    dev = "cuda:0"
    shapes = [
        ("square-4k", 4096, 4096, 4096),
        ("bert-qkv", 4096, 2304, 768),
        ("bert-fc1", 4096, 3072, 768),
        ("bert-fc2-dgrad", 4096, 768, 3072),
        ("llama-gate", 2048, 14336, 4096),
        ("llama-down", 2048, 4096, 14336),
        ("wgrad-768", 768, 2304, 4096),
        # llama-lora bench shapes (bs16 x S512 -> M=8192)
        ("llama-q16", 8192, 4096, 4096),
        ("llama-kv16", 8192, 1024, 4096),
        ("llama-gate16", 8192, 14336, 4096),
        ("llama-down16", 8192, 4096, 14336),
        ("llama-head16", 8192, 128256, 4096),
        # BERT bs128 wgrad shapes (split-K NT: dW = dy_t @ x, K = tokens)
        ("bert-wg-qkv", 2304, 768, 16384),
        ("bert-wg-fc1", 3072, 768, 16384),
        ("bert-wg-fc2", 768, 3072, 16384),
    ]
    print(f"{'shape':<16}{'M':>6}{'N':>7}{'K':>7}  {'NT TF':>8}{'NN TF':>8}{'TN TF':>8}")
    for name, M, N, K in shapes:
        torch.manual_seed(0)
        flops = 2.0 * M * N * K
        A_nt = (torch.rand(M, K, device=dev, dtype=torch.bfloat16) * 2 - 1).contiguous()
        B_nt = (torch.rand(N, K, device=dev, dtype=torch.bfloat16) * 2 - 1).contiguous()
        t_nt = bench(lambda: ops.gemm(A_nt, B_nt, 0))
        B_nn = (torch.rand(K, N, device=dev, dtype=torch.bfloat16) * 2 - 1).contiguous()
        t_nn = bench(lambda: ops.gemm(A_nt, B_nn, 1))
        A_tn = (torch.rand(K, M, device=dev, dtype=torch.bfloat16) * 2 - 1).contiguous()
        t_tn = bench(lambda: ops.gemm(A_tn, B_nn, 2))
        print(f"{name:<16}{M:>6}{N:>7}{K:>7}  {flops/t_nt/1e12:>8.0f}{flops/t_nn/1e12:>8.0f}{flops/t_tn/1e12:>8.0f}")


if __name__ == "__main__":
    main()
