"""Per-shape conv microbenchmark: the hand-written CDNA4 implicit-GEMM
kernels vs the PyTorch-ROCm (MIOpen) path on every WaterNet conv shape.

Backs docs/KERNELS.md's design claim ("general convolution belongs to
MIOpen; these kernels exist to beat it on THIS model") with numbers.

What is timed per shape, steady state (50 iters after 10 warmup):
  native : conv_bias_act on NHWC bf16 — fused conv+bias+ReLU/Sigmoid,
           packed weights cached (as in the training step)
  miopen-ncl / miopen-cl : F.conv2d(bf16) + bias + activation on NCHW /
           channels_last, torch.backends.cudnn.benchmark=True so MIOpen
           picks its best algorithm (find mode) during warmup

Run (GPU box):  python tools/conv_microbench.py [out.json]
"""

import json
import sys
import time
from pathlib import Path

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

import torch
import torch.nn.functional as F

# (name, Cin, Cout, ks, act) for bs=16 @ 112x112 — net.py:14-45,66-71
WATERNET_SHAPES = [
    ("cmg.conv1", 12, 128, 7, "relu"),
    ("cmg.conv2", 128, 128, 5, "relu"),
    ("cmg.conv3", 128, 128, 3, "relu"),
    ("cmg.conv4", 128, 64, 1, "relu"),
    ("cmg.conv5", 64, 64, 7, "relu"),
    ("cmg.conv6", 64, 64, 5, "relu"),
    ("cmg.conv7", 64, 64, 3, "relu"),
    ("cmg.conv8", 64, 3, 3, "sigmoid"),
    ("refiner.conv1", 6, 32, 7, "relu"),
    ("refiner.conv2", 32, 32, 5, "relu"),
    ("refiner.conv3", 32, 3, 3, "relu"),
]
# a few VGG tower shapes (112in): deep layers are the split-K territory
VGG_SHAPES = [
    ("vgg 112sq 64->64", 64, 64, 3, "relu", 112),
    ("vgg 56sq 128->128", 128, 128, 3, "relu", 56),
    ("vgg 28sq 256->256", 256, 256, 3, "relu", 28),
    ("vgg 14sq 512->512", 512, 512, 3, "relu", 14),
    ("vgg 7sq 512->512", 512, 512, 3, "relu", 7),
]

BS, H, W = 16, 112, 112
WARMUP, ITERS = 10, 50


def _time(fn):
    for _ in range(WARMUP):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(ITERS):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / ITERS * 1e6  # us


def bench_shape(name, cin, cout, ks, act, h=H, w=W):
    from waternet_amd.ops.conv import (
        ACT_RELU,
        ACT_SIGMOID,
        ConvSpec,
        conv_bias_act,
    )

    torch.manual_seed(0)
    mod = torch.nn.Conv2d(cin, cout, ks, padding="same").cuda()
    spec = ConvSpec(mod, ACT_RELU if act == "relu" else ACT_SIGMOID)
    cp = max(16, 1 << (cin - 1).bit_length())
    x_nhwc = torch.randn(BS, h, w, cp, device="cuda").bfloat16()
    if cp > cin:
        x_nhwc[..., cin:] = 0  # pad channels zero by construction
    native_us = _time(lambda: conv_bias_act(x_nhwc, spec))

    actf = torch.relu if act == "relu" else torch.sigmoid
    wt = mod.weight.detach().bfloat16()
    bi = mod.bias.detach().bfloat16()
    res = {"shape": f"{cin}->{cout} k{ks} {h}x{w}", "name": name,
           "native_us": round(native_us, 1)}
    for tag, mf in (("miopen_ncl_us", torch.contiguous_format),
                    ("miopen_cl_us", torch.channels_last)):
        try:
            x = torch.randn(BS, cin, h, w, device="cuda").bfloat16() \
                .to(memory_format=mf)
            wmf = wt.to(memory_format=mf)
            us = _time(lambda: actf(
                F.conv2d(x, wmf, bi, padding=ks // 2)))
            res[tag] = round(us, 1)
        except Exception as e:  # noqa: BLE001
            res[tag] = f"error: {e}"
    best = min(v for k, v in res.items()
               if k.endswith("_us") and isinstance(v, float)
               and k != "native_us")
    res["speedup_vs_best_miopen"] = round(best / native_us, 2)
    return res


def main():
    assert torch.cuda.is_available()
    torch.backends.cudnn.benchmark = True  # let MIOpen find its best algo
    rows = []
    for name, cin, cout, ks, act in WATERNET_SHAPES:
        rows.append(bench_shape(name, cin, cout, ks, act))
        print(rows[-1])
    for name, cin, cout, ks, act, hw in VGG_SHAPES:
        rows.append(bench_shape(name, cin, cout, ks, act, hw, hw))
        print(rows[-1])
    out = sys.argv[1] if len(sys.argv) > 1 else "conv_microbench.json"
    with open(out, "w") as f:
        json.dump({"bs": BS, "iters": ITERS, "rows": rows}, f, indent=1)
    geo = 1.0
    n = 0
    for r in rows:
        if isinstance(r["speedup_vs_best_miopen"], float):
            geo *= r["speedup_vs_best_miopen"]
            n += 1
    print(f"geomean speedup vs best MIOpen layout: {geo ** (1 / n):.2f}x "
          f"over {n} shapes -> {out}")


if __name__ == "__main__":
    main()
