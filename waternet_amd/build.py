"""Build helper: `python -m waternet_amd.build` compiles the HIP extension
in-tree (waternet_amd/_C*.so) for gfx950."""

import os
import subprocess
import sys
from pathlib import Path

REPO = Path(__file__).resolve().parent.parent


def build(verbose: bool = True) -> None:
    env = dict(os.environ)
    env.setdefault("PYTORCH_ROCM_ARCH", "gfx950")
    env.setdefault("MAX_JOBS", str(min(16, os.cpu_count() or 8)))
    cmd = [sys.executable, "setup.py", "build_ext", "--inplace"]
    res = subprocess.run(cmd, cwd=REPO, env=env, capture_output=not verbose,
                         text=True)
    if res.returncode != 0:
        out = (res.stdout or "") + (res.stderr or "")
        raise RuntimeError(f"native build failed:\n{out[-8000:]}")


if __name__ == "__main__":
    build()
