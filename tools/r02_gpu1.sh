#!/bin/bash
# Round-2 GPU validation call 1
set -x
mkdir -p gpurun_out
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -x -q 2>&1 | tail -25 > gpurun_out/r02_pytest.log
timeout 300 python bench.py --steps 30 --warmup 10 > gpurun_out/r02_bench.log 2>&1
timeout 300 python bench.py --mode infer1080p --steps 100 --warmup 10 > gpurun_out/r02_infer.log 2>&1
# train.py fast-path throughput: 2 epochs synthetic-800 (720 train/80 val per epoch)
WATERNET_TRAINING_DIR=/tmp/tr timeout 600 python train.py --synthetic 800 --epochs 3 --batch-size 16 --height 112 --width 112 > gpurun_out/r02_train_cli.log 2>&1
# rocprof kernel stats of the bench step (verify zero nchw<->nhwc bridges)
cd /tmp && export TMPDIR=/tmp && cd /root/repo
timeout 420 rocprofv3 --kernel-trace --stats -d gpurun_out/prof_r02 -o r02_step -- python bench.py --steps 5 --warmup 3 > gpurun_out/r02_prof.log 2>&1
echo DONE
