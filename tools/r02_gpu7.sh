#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q --tb=line > gpurun_out/r07_pytest.log 2>&1
# CLI evidence refresh (train fast, score, inference on GPU)
WATERNET_TRAINING_DIR=/tmp/tr timeout 600 python train.py --synthetic 200 --epochs 2 --batch-size 16 --height 112 --width 112 --full-state > gpurun_out/r07_cli_train.log 2>&1
timeout 300 python score.py --weights /tmp/tr/0/last.pt --synthetic 200 > gpurun_out/r07_cli_score.log 2>&1
python - << 'PYEOF' > gpurun_out/r07_cli_infer.log 2>&1
import numpy as np
from PIL import Image
rng = np.random.default_rng(0)
Image.fromarray(rng.integers(0,256,size=(256,320,3),dtype=np.uint8)).save("/tmp/frame.png")
PYEOF
timeout 300 python inference.py --source /tmp/frame.png --weights /tmp/tr/0/last.pt --name r07 >> gpurun_out/r07_cli_infer.log 2>&1
ls output/r07 >> gpurun_out/r07_cli_infer.log 2>&1
# 256-square config record
timeout 300 python bench.py --steps 15 --warmup 5 --height 256 --width 256 > gpurun_out/r07_b256.log 2>&1
# endurance soak
timeout 600 python tools/endurance.py > gpurun_out/r07_endurance.log 2>&1
# final flagship bench x2
timeout 300 python bench.py --steps 50 --warmup 15 > gpurun_out/r07_bench.log 2>&1
echo DONE
