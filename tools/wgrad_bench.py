"""Per-shape wgrad kernel timing (A/B between kernel generations via
WN_WGRAD_V4=0|1). Shapes = the WaterNet training layers at bs=16 112^2."""

import os
import sys
import time
from pathlib import Path

import torch

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from waternet_amd.ops import ext  # noqa: E402
from waternet_amd.ops.conv import pow2_channels  # noqa: E402

SHAPES = [  # (ks, C, K, label)
    (7, 12, 128, "cmg.conv1"),
    (5, 128, 128, "cmg.conv2"),
    (3, 128, 128, "cmg.conv3"),
    (1, 128, 64, "cmg.conv4"),
    (7, 64, 64, "cmg.conv5"),
    (5, 64, 64, "cmg.conv6"),
    (3, 64, 64, "cmg.conv7"),
    (7, 6, 32, "refiner.conv1(x3)"),
    (5, 32, 32, "refiner.conv2(x3)"),
]


def main():
    e = ext()
    N = int(os.environ.get("WG_N", "16"))
    H = W = int(os.environ.get("WG_HW", "112"))
    torch.manual_seed(0)
    total = 0.0
    print(f"WN_WGRAD_V4={os.environ.get('WN_WGRAD_V4', '(default 1)')} "
          f"N={N} HW={H}")
    for ks, C, K, label in SHAPES:
        Cp, Kp = pow2_channels(C), pow2_channels(K)
        dy = torch.randn(N, H, W, Kp, device="cuda",
                         dtype=torch.bfloat16).contiguous()
        x = torch.randn(N, H, W, Cp, device="cuda",
                        dtype=torch.bfloat16).contiguous()
        dw = torch.zeros(K, C, ks, ks, device="cuda", dtype=torch.float32)
        for _ in range(3):
            e.conv2d_wgrad(dy, x, dw, ks)
        torch.cuda.synchronize()
        t0 = time.perf_counter()
        iters = 20
        for _ in range(iters):
            e.conv2d_wgrad(dy, x, dw, ks)
        torch.cuda.synchronize()
        us = (time.perf_counter() - t0) / iters * 1e6
        mult = 3 if "x3" in label else 1
        total += us * mult
        flops = 2.0 * N * H * W * K * C * ks * ks
        print(f"  {label:20s} ks{ks} {C:3d}->{K:3d}: {us:7.1f} us "
              f"({flops / us / 1e6:6.1f} TF/s)")
    print(f"  TOTAL (per step, x3 refiners counted): {total:.0f} us")


if __name__ == "__main__":
    main()
