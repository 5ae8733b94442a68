#!/bin/bash
set -x
mkdir -p gpurun_out profiles
export HSA_ENABLE_IPC_MODE_LEGACY=0
python -m pytest tests/test_gpu_engine.py tests/test_gpu_ddp.py -q --tb=line > gpurun_out/r08_pytest.log 2>&1
# odd-size CLI smoke through the padded GPU engine
python - << 'PYEOF' > gpurun_out/r08_oddsize.log 2>&1
import numpy as np
from PIL import Image
rng = np.random.default_rng(1)
Image.fromarray(rng.integers(0,256,size=(45,67,3),dtype=np.uint8)).save("/tmp/odd.png")
PYEOF
timeout 300 python inference.py --source /tmp/odd.png --name r08odd >> gpurun_out/r08_oddsize.log 2>&1
python - << 'PYEOF' >> gpurun_out/r08_oddsize.log 2>&1
import numpy as np
from PIL import Image
out = np.asarray(Image.open("output/r08odd/odd.png"))
print("odd-size output:", out.shape, out.dtype)
PYEOF
# PMC duty table for the current 112^2 step
cd /tmp && export TMPDIR=/tmp && cd "$GRAFT_REPO_ROOT"
timeout 420 rocprofv3 --pmc SQ_WAVE_CYCLES,SQ_WAIT_INST_ANY,SQ_WAIT_ANY,SQ_VALU_MFMA_BUSY_CYCLES --output-format csv -d gpurun_out/pmc112 -o p112 -- python bench.py --steps 3 --warmup 2 --no-graph > gpurun_out/r08_pmc112.log 2>&1
# convergence (tail-summary version)
timeout 900 python tools/convergence_check.py --flagship > gpurun_out/r08_convergence.log 2>&1
cp profiles/convergence_flagship.json gpurun_out/ 2>/dev/null
echo DONE
