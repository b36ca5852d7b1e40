#!/bin/bash
# Same-box A/B of wgrad accumulation flavor + new delta kernel check.
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out

echo "=== gpu attention tests (delta kernel changed) ==="
timeout 600 python -m pytest tests/test_gpu_attention.py tests/test_gpu_kernels.py -m gpu -q 2>&1 | tail -2

echo "=== kernels_micro (delta bandwidth) ==="
timeout 300 python benchmarks/kernels_micro.py 2>&1 | grep -E "delta|adamw"

echo "=== bench A: wgrad add (default) ==="
ACCO_WGRAD_EPILOGUE=0 timeout 420 python bench.py --steps 20 --warmup 4 2>/dev/null | tail -1
echo "=== bench B: wgrad epilogue beta=1 ==="
ACCO_WGRAD_EPILOGUE=1 timeout 420 python bench.py --steps 20 --warmup 4 2>/dev/null | tail -1
echo "=== bench A again (drift check) ==="
ACCO_WGRAD_EPILOGUE=0 timeout 420 python bench.py --steps 20 --warmup 4 2>/dev/null | tail -1
echo R1_EPILOGUE_AB_DONE
