#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
# end-to-end product demo: 100 epochs through the CLI at engine speed
WATERNET_TRAINING_DIR=/tmp/tr timeout 900 python train.py --synthetic 800 --epochs 100 --batch-size 16 --height 112 --width 112 --full-state > gpurun_out/r10_train100.log 2>&1
mkdir -p gpurun_out/demo_run
cp /tmp/tr/0/metrics-train.csv /tmp/tr/0/metrics-val.csv /tmp/tr/0/config.json gpurun_out/demo_run/ 2>/dev/null
timeout 300 python score.py --weights /tmp/tr/0/last.pt --synthetic 800 > gpurun_out/r10_score.log 2>&1
# resume check: 2 more epochs from the sidecar
WATERNET_TRAINING_DIR=/tmp/tr timeout 300 python train.py --synthetic 800 --epochs 102 --batch-size 16 --height 112 --width 112 --resume /tmp/tr/0 > gpurun_out/r10_resume.log 2>&1
# 2000-step soak (fixed invocation)
python - << 'PYEOF' > gpurun_out/r10_soak2000.log 2>&1
import sys, pathlib, time, torch
sys.path.insert(0, str(pathlib.Path(".").resolve()))
from waternet_amd.engine.fast import BenchTrainer
tr = BenchTrainer(batch_size=16, height=112, width=112, device="cuda:0", use_graph=True, seed=5)
for _ in range(20): tr.step()
torch.cuda.synchronize(); m0 = torch.cuda.memory_allocated(); t0 = time.perf_counter()
for _ in range(2000): tr.step()
torch.cuda.synchronize(); dt = time.perf_counter()-t0; m1 = torch.cuda.memory_allocated()
print(f"img/s={2000*16/dt:.1f} mem0={m0/1e6:.1f}MB mem1={m1/1e6:.1f}MB peak={torch.cuda.max_memory_allocated()/1e9:.2f}GB")
assert m1 <= m0 + 2**20, "memory growth"
print("SOAK2000 OK")
PYEOF
echo DONE
