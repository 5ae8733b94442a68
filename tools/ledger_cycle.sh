#!/bin/bash
# One evidence cycle on a GPU box: suite subset + bench sample with clock
# context appended to gpurun_out/ledger_entry.json
set -x
export HSA_ENABLE_IPC_MODE_LEGACY=0
SUITES=${@:-tests/test_gpu_engine.py}
python -m pytest $SUITES -q 2>&1 | grep -E "passed|failed"
timeout 300 python bench.py --steps 30 --warmup 10 2>/dev/null | tail -1 > /tmp/b.json
SCLK=$(rocm-smi --showgpuclocks 2>/dev/null | grep -oE '\([0-9]+Mhz\)' | head -1 | tr -d '(Mhz)')
python3 - << PYEOF
import json
d = json.load(open("/tmp/b.json"))
d["sclk_mhz_after"] = int("${SCLK:-0}") or None
open("gpurun_out/ledger_entry.json", "w").write(json.dumps(d) + "\n")
print(round(d["value"], 1), "img/s, sclk", d["sclk_mhz_after"])
PYEOF
