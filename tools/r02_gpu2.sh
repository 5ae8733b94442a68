#!/bin/bash
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q --tb=line > gpurun_out/r02b_pytest.log 2>&1
timeout 300 python -m pytest "tests/test_gpu_engine.py::test_train_cli_fast_gpu" -q --tb=long > gpurun_out/r02b_clitest.log 2>&1
timeout 240 python tools/wgrad_bench.py > gpurun_out/r02b_wgrad_v4.log 2>&1
WN_WGRAD_V4=0 timeout 240 python tools/wgrad_bench.py > gpurun_out/r02b_wgrad_v3.log 2>&1
timeout 300 python bench.py --steps 30 --warmup 10 > gpurun_out/r02b_bench.log 2>&1
WATERNET_TRAINING_DIR=/tmp/tr timeout 600 python train.py --synthetic 800 --epochs 3 --batch-size 16 --height 112 --width 112 > gpurun_out/r02b_train_cli.log 2>&1
echo DONE
