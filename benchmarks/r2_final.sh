#!/bin/bash
# Round-2 final same-box measurement sweep (one gpurun box, one pass):
# GPU test suite, method matrix, model family benches, kernel micros and a
# 300-step soak. Output is copied to profiles/r02_final_measurements.txt.
set -u
cd "${GRAFT_REPO_ROOT:-/root/repo}"
OUT=gpurun_out/r2_final.txt
{
  echo "# Round-2 final measurements (one box, one pass; $(date -u +%Y-%m-%dT%H:%MZ))"
  echo "## GPU test suite"
  python -m pytest tests -m gpu -q 2>&1 | tail -2
  echo
  echo "## llama-1b method matrix (b8 s1024)"
  for m in acco ddp dpu; do
    v=$(timeout 300 python bench.py --gpus 1 --steps 15 --warmup 4 --method $m 2>/dev/null | tail -1)
    echo "$m: $v"
  done
  echo
  echo "## gptneo-125m (reference headline model)"
  timeout 300 python bench.py --gpus 1 --steps 15 --warmup 4 --model gptneo-125m 2>/dev/null | tail -1
  echo
  echo "## llama-8b (BASELINE config 4 sizing)"
  timeout 420 python bench.py --gpus 1 --steps 5 --warmup 2 --model llama-8b --batch 4 --seq 512 2>/dev/null | tail -1
  timeout 420 python bench.py --gpus 1 --steps 5 --warmup 2 --model llama-8b --batch 8 --seq 1024 2>/dev/null | tail -1
  echo
  echo "## attention micro (llama-1b shape / 8B shape)"
  python benchmarks/attn_micro.py 2>/dev/null
  python benchmarks/attn_micro.py --8b 2>/dev/null
  echo
  echo "## memory-bound kernel micro"
  python benchmarks/kernels_micro.py 2>/dev/null
  echo
  echo "## 300-step soak (llama-1b acco)"
  timeout 600 python bench.py --gpus 1 --steps 300 --warmup 10 2>/dev/null | tail -1
} > "$OUT" 2>&1
echo "R2_FINAL_DONE rc=$?"
tail -3 "$OUT"
