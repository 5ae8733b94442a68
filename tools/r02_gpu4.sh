#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
WATERNET_TRAINING_DIR=/tmp/tr timeout 600 python train.py --synthetic 800 --epochs 4 --batch-size 16 --height 112 --width 112 > gpurun_out/r04_train_cli.log 2>&1
timeout 420 python tools/infer_stability.py > gpurun_out/r04_stability.log 2>&1
cp profiles/infer_stability.json gpurun_out/ 2>/dev/null
timeout 900 python tools/convergence_check.py --flagship > gpurun_out/r04_convergence.log 2>&1
cp profiles/convergence_flagship.json gpurun_out/ 2>/dev/null
timeout 300 python bench.py --steps 8 --warmup 3 --batch-size 64 --height 512 --width 512 > gpurun_out/r04_b512.log 2>&1
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof112 -o s112 -- python bench.py --steps 5 --warmup 3 > gpurun_out/r04_prof112.log 2>&1
timeout 420 rocprofv3 --kernel-trace --stats --output-format csv -d gpurun_out/prof512 -o s512 -- python bench.py --steps 3 --warmup 2 --batch-size 64 --height 512 --width 512 --no-graph > gpurun_out/r04_prof512.log 2>&1
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_INST_ANY,SQ_WAIT_ANY,SQ_VALU_MFMA_BUSY_CYCLES --output-format csv -d gpurun_out/pmc512 -o p512 -- python bench.py --steps 2 --warmup 1 --batch-size 64 --height 512 --width 512 --no-graph > gpurun_out/r04_pmc512.log 2>&1
timeout 420 rocprofv3 --pmc FETCH_SIZE --output-format csv -d gpurun_out/tcc512 -o t512 -- python bench.py --steps 2 --warmup 1 --batch-size 64 --height 512 --width 512 --no-graph > gpurun_out/r04_tcc512.log 2>&1
echo DONE
