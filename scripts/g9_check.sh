#!/bin/bash
# g9 parity-fix validation: refchecks + the two GEMM-sensitive benches.
set -u
timeout 200 python scripts/debug_g9split.py 2>&1 | tail -n 8
timeout 200 python -m pytest tests/test_kernels_gpu.py -x -q \
  -k "gemm_router or 8phase" 2>&1 | tail -n 2
timeout 300 python bench.py --model bert-base --steps 12 --warmup 4 2>&1 | tail -n 1
timeout 300 python bench.py --model llama-lora --steps 12 --warmup 4 2>&1 | tail -n 1
