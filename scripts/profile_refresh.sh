#!/bin/bash
# Refresh bert/llama kernel breakdowns after the g9 fix + LoRA-join fusion.
set -u
export TMPDIR=/tmp
cd "$GRAFT_REPO_ROOT"
rm -rf gpurun_out/prof_bert_fixed gpurun_out/prof_llama_fixed
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_bert_fixed -- \
  python bench.py --model bert-base --steps 6 --warmup 2 > /dev/null 2>&1
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_llama_fixed -- \
  python bench.py --model llama-lora --steps 6 --warmup 2 > /dev/null 2>&1
python benchmarks/profstats.py gpurun_out/prof_bert_fixed 2>&1 | head -n 24 \
  | tee gpurun_out/bert_fixed_stats.txt
python benchmarks/profstats.py gpurun_out/prof_llama_fixed 2>&1 | head -n 24 \
  | tee gpurun_out/llama_fixed_stats.txt
# keep only the text summaries (db files are big)
rm -rf gpurun_out/prof_bert_fixed gpurun_out/prof_llama_fixed
