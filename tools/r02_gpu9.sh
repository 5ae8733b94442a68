#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
# convergence A/B at the reference's 256^2 headline quality config
timeout 1200 python tools/convergence_check.py --steps 400 --bs 16 --hw 256 --out convergence_256sq.json > gpurun_out/r09_conv256.log 2>&1
cp profiles/convergence_256sq.json gpurun_out/ 2>/dev/null
# 2000-step soak
sed 's/range(300)/range(2000)/' tools/endurance.py > /tmp/endurance2000.py
timeout 900 python /tmp/endurance2000.py > gpurun_out/r09_soak2000.log 2>&1
# bench distribution: 5 back-to-back
for i in 1 2 3 4 5; do
  timeout 300 python bench.py --steps 30 --warmup 10 2>/dev/null | tail -1 >> gpurun_out/r09_bench5.log
done
echo DONE
