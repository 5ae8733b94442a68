#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests -m gpu -q --tb=line > gpurun_out/r06_pytest.log 2>&1
timeout 900 python tools/convergence_check.py --flagship > gpurun_out/r06_convergence.log 2>&1
cp profiles/convergence_flagship.json gpurun_out/ 2>/dev/null
timeout 420 python tools/infer_stability.py > gpurun_out/r06_stability.log 2>&1
cp profiles/infer_stability.json gpurun_out/ 2>/dev/null
timeout 300 python -m torch.distributed.run --standalone --local-addr 127.0.0.1 --nproc-per-node 1 bench.py --gpus 1 --steps 10 --warmup 5 > gpurun_out/r06_torchrun.log 2>&1
timeout 300 python bench.py --steps 30 --warmup 10 > gpurun_out/r06_bench.log 2>&1
echo DONE
