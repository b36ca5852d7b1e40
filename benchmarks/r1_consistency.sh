#!/bin/bash
# Round-1 closing evidence on one box with the final code: method/shape/model
# sweeps, convergence re-check, 100-step soak, final kernel trace.
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out
B() { timeout 420 python bench.py "$@" 2>/dev/null | tail -1; }

echo "=== methods (llama-1b) ==="
echo -n "acco "; B --steps 10 --warmup 3
echo -n "dpu  "; B --method dpu --steps 10 --warmup 3
echo -n "ddp  "; B --method ddp --steps 10 --warmup 3

echo "=== shapes (acco) ==="
echo -n "b16   "; B --steps 10 --warmup 3 --batch 16
echo -n "b32   "; B --steps 10 --warmup 3 --batch 32
echo -n "s2048 "; B --steps 10 --warmup 3 --batch 4 --seq 2048
echo -n "s4096 "; B --steps 10 --warmup 3 --batch 2 --seq 4096

echo "=== models ==="
echo -n "neo   "; B --model gptneo-125m --steps 10 --warmup 3
echo -n "8b    "; B --model llama-8b --batch 4 --seq 512 --steps 10 --warmup 3

echo "=== soak 100 steps ==="
B --steps 100 --warmup 5

echo "=== train evidence (llama, 300 steps) ==="
timeout 900 python benchmarks/train_evidence.py 300 --llama 2>&1 | tail -8

echo "=== prof8 kernel trace ==="
export TMPDIR=/tmp
(cd /tmp && timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof8" -o prof8 -- \
    bash -c "cd $GRAFT_REPO_ROOT && python bench.py --steps 3 --warmup 2" \
    > /dev/null 2> "$GRAFT_REPO_ROOT/gpurun_out/prof8.err")
echo "trace rc=$?"
echo R1_CONSISTENCY_DONE
