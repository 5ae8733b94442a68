#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
# The reference's FULL training duration: 400 epochs, 800 images, bs=16,
# 112x112 (reference wall: 28,182.8 s). Synthetic stand-in dataset.
WATERNET_TRAINING_DIR=/tmp/tr timeout 1500 python train.py --synthetic 800 --epochs 400 --batch-size 16 --height 112 --width 112 --full-state > gpurun_out/r20_train400.log 2>&1
mkdir -p gpurun_out/demo400
cp /tmp/tr/0/metrics-train.csv /tmp/tr/0/metrics-val.csv /tmp/tr/0/config.json gpurun_out/demo400/ 2>/dev/null
timeout 300 python score.py --weights /tmp/tr/0/last.pt --synthetic 800 > gpurun_out/r20_score.log 2>&1
# 30k-step soak
python - << 'PYEOF' > gpurun_out/r20_soak30k.log 2>&1
import sys, pathlib, time, torch
sys.path.insert(0, str(pathlib.Path(".").resolve()))
from waternet_amd.engine.fast import BenchTrainer
tr = BenchTrainer(batch_size=16, height=112, width=112, device="cuda:0", use_graph=True, seed=5)
for _ in range(20): tr.step()
torch.cuda.synchronize(); m0 = torch.cuda.memory_allocated(); t0 = time.perf_counter()
for i in range(30000):
    tr.step()
    if (i+1) % 10000 == 0:
        torch.cuda.synchronize()
        print(f"step {i+1}: {(i+1)*16/(time.perf_counter()-t0):.1f} img/s cum", flush=True)
torch.cuda.synchronize(); dt = time.perf_counter()-t0; m1 = torch.cuda.memory_allocated()
import math
m = tr.metrics()
assert all(math.isfinite(v) for v in m.values()), m
assert m1 <= m0 + 2**20
print(f"SOAK30K OK img/s={30000*16/dt:.1f} mem flat={m0==m1} peak={torch.cuda.max_memory_allocated()/1e9:.2f}GB")
PYEOF
echo DONE
