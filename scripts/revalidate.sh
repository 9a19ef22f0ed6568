#!/bin/bash
# Round-2 revalidation: full GPU suite, smoke, and the four benches.
set -u
timeout 680 python -m pytest tests -m gpu -q 2>&1 | tail -n 4
timeout 180 python -c 'import __graft_entry__ as g; g.smoke(); print("SMOKE-OK")' 2>&1 | tail -n 2
for m in resnet18 resnet50 bert-base llama-lora; do
  timeout 300 python bench.py --model "$m" --steps 12 --warmup 4 2>&1 | tail -n 1
done
