"""1080p inference stability evidence: N in-process
runs of the hipGraph-captured per-frame pipeline with GPU clock sampling
(rocm-smi) alongside, so run-to-run fps spread can be attributed (DVFS vs
code). Writes profiles/infer_stability.json."""

import json
import pathlib
import re
import subprocess
import sys
import threading
import time

sys.path.insert(0, str(pathlib.Path(__file__).resolve().parent.parent))

import numpy as np
import torch

from waternet_amd.engine.inferencer import InferenceEngine
from waternet_amd.models.waternet import WaterNet

H, W = 1088, 1920
RUNS = 5
FRAMES = 200


class ClockSampler(threading.Thread):
    def __init__(self, period=0.1):
        super().__init__(daemon=True)
        self.period = period
        self.samples = []
        self._halt = threading.Event()

    def run(self):
        while not self._halt.is_set():
            try:
                out = subprocess.run(
                    ["rocm-smi", "--showgpuclocks", "--showpower",
                     "--showtemp"],
                    capture_output=True, text=True, timeout=5,
                ).stdout
                mhz = re.search(r"sclk.*?\((\d+)Mhz\)", out)
                watts = re.search(r"Power \(W\): (\d+\.?\d*)", out)
                temp = re.search(r"Temperature.*?(\d+\.?\d*)c", out, re.I)
                self.samples.append({
                    "t": time.time(),
                    "sclk_mhz": int(mhz.group(1)) if mhz else None,
                    "power_w": float(watts.group(1)) if watts else None,
                    "temp_c": float(temp.group(1)) if temp else None,
                })
            except Exception:  # noqa: BLE001
                pass
            time.sleep(self.period)

    def stop(self):
        self._halt.set()


def main():
    torch.manual_seed(0)
    model = WaterNet().to("cuda:0")
    eng = InferenceEngine(model, H, W, device="cuda:0", use_graph=True)
    rng = np.random.default_rng(0)
    frames = [rng.integers(0, 256, size=(H, W, 3), dtype=np.uint8)
              for _ in range(4)]
    for i in range(10):
        eng.infer_frame(frames[i % 4])
    torch.cuda.synchronize()

    sampler = ClockSampler()
    sampler.start()
    runs = []
    for r in range(RUNS):
        t0 = time.perf_counter()
        for i in range(FRAMES):
            eng.infer_frame(frames[i % 4])
        torch.cuda.synchronize()
        dt = time.perf_counter() - t0
        fps = FRAMES / dt
        runs.append({"run": r, "fps": fps, "t0": t0,
                     "t1": time.perf_counter()})
        print(f"run {r}: {fps:.1f} fps")
        time.sleep(0.5)
    sampler.stop()
    sampler.join(timeout=2)

    fpss = sorted(x["fps"] for x in runs)
    clocks = [s["sclk_mhz"] for s in sampler.samples if s["sclk_mhz"]]
    result = {
        "shape": f"{H}x{W}", "frames_per_run": FRAMES, "runs": runs,
        "fps_p50": fpss[len(fpss) // 2], "fps_min": fpss[0],
        "fps_max": fpss[-1],
        "spread_pct": 100.0 * (fpss[-1] - fpss[0]) / fpss[-1],
        "clock_samples": sampler.samples,
        "clock_mhz_minmax": [min(clocks), max(clocks)] if clocks else None,
    }
    out = (pathlib.Path(__file__).resolve().parent.parent / "profiles"
           / "infer_stability.json")
    out.write_text(json.dumps(result, indent=1))
    print(json.dumps({k: v for k, v in result.items()
                      if k not in ("runs", "clock_samples")}))


if __name__ == "__main__":
    main()
