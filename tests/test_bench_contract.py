"""Driver-contract guards: bench.py's flags and its recorded JSON line
must keep the exact schema the driver parses (BASELINE metric/config)."""

import json
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent

REQUIRED_KEYS = {
    "metric", "value", "unit", "n_gpus", "steps", "warmup", "ms_per_step",
    "higher_is_better", "scaling", "vs_baseline", "dtype", "data", "config",
}


def test_bench_args_defaults():
    import sys

    import bench

    argv = sys.argv
    sys.argv = ["bench.py"]  # no flags: driver default invocation
    try:
        a = bench.parse_args()
    finally:
        sys.argv = argv
    assert a.gpus == 1 and a.steps == 30 and a.warmup == 10
    assert a.batch_size == 16 and a.height == 112 and a.width == 112
    assert a.mode == "train"


def test_recorded_bench_line_schema():
    rec = json.loads((REPO / "profiles" / "r02_bench_final.json").read_text())
    assert REQUIRED_KEYS <= set(rec.keys()), REQUIRED_KEYS - set(rec.keys())
    assert rec["metric"].startswith("train images/sec")
    assert rec["higher_is_better"] is True
    assert rec["scaling"] == "weak"
    assert rec["dtype"] == "bf16"
    assert rec["data"] == "synthetic"
    assert rec["config"]["model"] == "waternet"
    assert rec["config"]["im_size"] == "112x112"
    assert rec["config"]["global_batch"] == rec["n_gpus"] * 16
    assert abs(rec["vs_baseline"] - rec["value"] / 12.8) < 1e-6
    # whole-job aggregate sanity: value == total images / elapsed
    assert abs(rec["value"] * rec["ms_per_step"] / 1000.0
               - rec["n_gpus"] * 16) < 1e-6


def test_bench_requires_gpu_clear_exit():
    """On a CPU-only box bench.py exits with a clear message (the driver
    only runs it on GPU boxes; anything else must not half-run)."""
    import subprocess
    import sys

    out = subprocess.run(
        [sys.executable, str(REPO / "bench.py"), "--steps", "1"],
        capture_output=True, text=True, timeout=600,
    )
    assert out.returncode != 0
    assert "requires a ROCm GPU" in out.stderr
