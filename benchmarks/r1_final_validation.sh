#!/bin/bash
# End-of-round validation: full GPU suite, smoke, headline trio, convergence
# re-check, final kernel trace — all on the final code.
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out

echo "=== full gpu suite ==="
timeout 900 python -m pytest tests -m gpu -q 2>&1 | tail -1
echo "=== smoke ==="
timeout 600 python __graft_entry__.py smoke 2>&1 | tail -1

echo "=== headline trio ==="
echo -n "acco "; timeout 420 python bench.py --steps 15 --warmup 4 2>/dev/null | tail -1
echo -n "dpu  "; timeout 420 python bench.py --method dpu --steps 10 --warmup 3 2>/dev/null | grep -oE '"value": [0-9.]+'
echo -n "ddp  "; timeout 420 python bench.py --method ddp --steps 10 --warmup 3 2>/dev/null | grep -oE '"value": [0-9.]+'
echo -n "neo  "; timeout 420 python bench.py --model gptneo-125m --steps 10 --warmup 3 2>/dev/null | grep -oE '"value": [0-9.]+'

echo "=== convergence re-check (200 steps, all final kernels) ==="
timeout 600 python benchmarks/train_evidence.py 200 --llama 2>&1 | grep "LOSS_CURVE" | python3 -c "
import sys, ast
for line in sys.stdin:
    pts = ast.literal_eval(line.split('LOSS_CURVE')[1].strip())
    print('first', pts[0], 'last', pts[-1])
"

echo "=== prof9 kernel trace ==="
export TMPDIR=/tmp
(cd /tmp && timeout 600 rocprofv3 --kernel-trace --stats -d "$GRAFT_REPO_ROOT/gpurun_out/prof9" -o prof9 -- \
    bash -c "cd $GRAFT_REPO_ROOT && python bench.py --steps 3 --warmup 2" \
    > /dev/null 2> "$GRAFT_REPO_ROOT/gpurun_out/prof9.err")
echo "trace rc=$?"
echo R1_FINAL_VALIDATION_DONE
