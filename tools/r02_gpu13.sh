#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
# 10k-step soak (~95 s of GPU compute): drift + leak + graph longevity
python - << 'PYEOF' > gpurun_out/r13_soak10k.log 2>&1
import sys, pathlib, time, torch
sys.path.insert(0, str(pathlib.Path(".").resolve()))
from waternet_amd.engine.fast import BenchTrainer
tr = BenchTrainer(batch_size=16, height=112, width=112, device="cuda:0", use_graph=True, seed=5)
for _ in range(20): tr.step()
torch.cuda.synchronize(); m0 = torch.cuda.memory_allocated(); t0 = time.perf_counter()
for i in range(10000):
    tr.step()
    if (i+1) % 2000 == 0:
        torch.cuda.synchronize()
        print(f"step {i+1}: {(i+1)*16/(time.perf_counter()-t0):.1f} img/s cum, mem={torch.cuda.memory_allocated()/1e6:.1f}MB", flush=True)
torch.cuda.synchronize(); dt = time.perf_counter()-t0; m1 = torch.cuda.memory_allocated()
m = tr.metrics()
print(f"FINAL img/s={10000*16/dt:.1f} mem0={m0/1e6:.1f} mem1={m1/1e6:.1f} peak={torch.cuda.max_memory_allocated()/1e9:.2f}GB")
import math
assert all(math.isfinite(v) for v in m.values()), m
assert m1 <= m0 + 2**20
print("SOAK10K OK", m)
PYEOF
# 2000-step flagship convergence A/B
timeout 1500 python tools/convergence_check.py --steps 2000 --bs 16 --hw 112 --out convergence_long.json > gpurun_out/r13_convlong.log 2>&1
cp profiles/convergence_long.json gpurun_out/ 2>/dev/null
echo DONE
