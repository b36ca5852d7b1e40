#!/bin/bash
# A/B: in-place dW accumulation (arena_linear/addmm_ beta=1) vs previous
# temp-dW + AccumulateGrad path, plus the PMC pass that failed last call.
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out

echo "=== gpu tests ==="
timeout 600 python -m pytest tests -m gpu -q 2>&1 | tail -2

echo "=== bench llama-1b acco (arena_linear) ==="
timeout 420 python bench.py --steps 10 --warmup 3 2>/dev/null | tee gpurun_out/bench_arena.json
echo "=== bench gptneo-125m acco ==="
timeout 420 python bench.py --model gptneo-125m --steps 10 --warmup 3 2>/dev/null | tee gpurun_out/bench_arena_neo.json
echo "=== bench llama-1b ddp ==="
timeout 420 python bench.py --method ddp --steps 10 --warmup 3 2>/dev/null | tee gpurun_out/bench_arena_ddp.json

echo "=== kernel trace (adds should shrink) ==="
export TMPDIR=/tmp
(cd /tmp && timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof7" -o prof7 -- \
    bash -c "cd $GRAFT_REPO_ROOT && python bench.py --steps 3 --warmup 2" \
    > /dev/null 2> "$GRAFT_REPO_ROOT/gpurun_out/prof7.err")
echo "trace rc=$?"

echo "=== PMC counters on kernels_micro ==="
(cd /tmp && timeout 600 rocprofv3 --pmc MfmaUtil VALUBusy MemUnitBusy LDSBankConflict \
    -d "$GRAFT_REPO_ROOT/gpurun_out/pmc_kernels" -o pmck -- \
    bash -c "cd $GRAFT_REPO_ROOT && python benchmarks/kernels_micro.py" \
    > /dev/null 2> "$GRAFT_REPO_ROOT/gpurun_out/pmc_kernels.err")
echo "PMC rc=$?"
echo R1_ARENA_AB_DONE
