#!/bin/bash
# Round-1 final GPU evidence: memory-bound kernel micro + PMC counters +
# TunableOp GEMM autotune A/B on the flagship bench.
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out

echo "=== kernels_micro (effective bandwidth) ==="
timeout 300 python benchmarks/kernels_micro.py 2>&1 | tee gpurun_out/kernels_micro.log

echo "=== kernels_micro PMC counters ==="
export TMPDIR=/tmp
(cd /tmp && timeout 600 rocprofv3 --pmc MfmaUtil VALUBusy MemUnitBusy LDSBankConflict \
    -d "$GRAFT_REPO_ROOT/gpurun_out/pmc_kernels" -o pmck -- \
    bash -c "cd $GRAFT_REPO_ROOT && python benchmarks/kernels_micro.py" \
    > /dev/null 2> gpurun_out/pmc_kernels.err)
echo "PMC rc=$?"

echo "=== bench baseline (no tunableop) ==="
timeout 420 python bench.py --steps 10 --warmup 3 2>/dev/null | tee gpurun_out/bench_notune.json

echo "=== tunableop tuning pass ==="
export PYTORCH_TUNABLEOP_ENABLED=1
export PYTORCH_TUNABLEOP_TUNING=1
export PYTORCH_TUNABLEOP_FILENAME="$GRAFT_REPO_ROOT/gpurun_out/tunableop_llama1b.csv"
export PYTORCH_TUNABLEOP_MAX_TUNING_DURATION_MS=50
export PYTORCH_TUNABLEOP_MAX_WARMUP_DURATION_MS=10
timeout 900 python bench.py --steps 3 --warmup 1 2>/dev/null | tail -1

echo "=== bench with tuned solutions ==="
export PYTORCH_TUNABLEOP_TUNING=0
timeout 420 python bench.py --steps 10 --warmup 3 2>/dev/null | tee gpurun_out/bench_tuned.json
echo R1_FINAL_GPU_DONE
