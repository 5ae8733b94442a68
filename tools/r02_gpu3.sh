#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q --tb=line > gpurun_out/r03_pytest.log 2>&1
timeout 300 python bench.py --steps 30 --warmup 10 > gpurun_out/r03_bench.log 2>&1
WATERNET_TRAINING_DIR=/tmp/tr timeout 600 python train.py --synthetic 800 --epochs 3 --batch-size 16 --height 112 --width 112 > gpurun_out/r03_train_cli.log 2>&1
timeout 420 python tools/infer_stability.py > gpurun_out/r03_stability.log 2>&1
cp profiles/infer_stability.json gpurun_out/ 2>/dev/null
timeout 900 python tools/convergence_check.py --flagship > gpurun_out/r03_convergence.log 2>&1
cp profiles/convergence_flagship.json gpurun_out/ 2>/dev/null
echo DONE
