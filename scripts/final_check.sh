#!/bin/bash
# Final round-2 validation: post-fix GEMM TF sweep + full GPU suite + smoke.
set -u
timeout 200 python benchmarks/g8_sched_ab.py 2>&1 | tail -n 8
timeout 680 python -m pytest tests -m gpu -q 2>&1 | tail -n 3
timeout 180 python -c 'import __graft_entry__ as g; g.smoke(); print("SMOKE-OK")' 2>&1 | tail -n 1
