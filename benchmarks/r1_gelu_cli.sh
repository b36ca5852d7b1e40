#!/bin/bash
# Fast-tanh gelu validation + gptneo bench + full CLI path on hardware.
set -x
cd "$GRAFT_REPO_ROOT" || exit 1
mkdir -p gpurun_out

echo "=== gpu kernel numerics (gelu changed) ==="
timeout 600 python -m pytest tests/test_gpu_kernels.py tests/test_gpu_model_parity.py -m gpu -q 2>&1 | tail -2

echo "=== gelu micro ==="
timeout 300 python benchmarks/kernels_micro.py 2>&1 | grep -E "gelu|layer_norm"

echo "=== gptneo bench (fast tanh) ==="
timeout 420 python bench.py --model gptneo-125m --steps 10 --warmup 3 2>/dev/null | tail -1

echo "=== main.py CLI end-to-end (llama-1b, 30 steps) ==="
timeout 600 python main.py train=acco data=synthetic model=llama-1b \
    train.nb_steps_tot=30 train.save=false run_name=gpucli 2>&1 | tail -4
ls -la results.csv tensorboard 2>/dev/null | head -5
echo "=== main.py CLI ddp (gptneo, 20 steps) ==="
timeout 600 python main.py train=ddp data=synthetic model=gptneo \
    train.nb_steps_tot=20 train.save=false run_name=gpucli2 2>&1 | tail -3
echo R1_GELU_CLI_DONE
