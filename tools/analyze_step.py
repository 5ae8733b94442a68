"""Cost-center analysis of a rocprof kernel-stats CSV.

Groups the kernel inventory of a profiled run into the pipeline stages
used throughout docs/HEADROOM.md (conv fwd/dgrad, wgrad, losses/metrics,
preprocess, optimizer, ...) and prints per-stage ms/step — the table that
drives optimization decisions.

    python tools/analyze_step.py profiles/r02f_train112_kernel_stats.csv 25
    (second arg = number of steps the profile covered)
"""

import csv
import re
import sys

STAGES = [
    ("conv fwd/dgrad (igemm)", r"k_conv_igemm"),
    ("conv wgrad", r"k_conv_wgrad|k_wgrad_smallk"),
    ("split-K finalize", r"k_splitk_finalize"),
    ("act/bias backward", r"k_act_bwd|k_abb_reduce|k_bias_grad"),
    ("losses/metrics", r"k_sqdiff255|k_ssim|k_normalize"),
    ("preprocess", r"k_wb_|k_rgb|k_clahe|k_gamma"),
    ("input/output bridges", r"k_build_inputs|k_u8_to|k_out_to_u8|"
                             r"k_nchw2nhwc|k_nhwc2nchw"),
    ("pooling", r"k_maxpool"),
    ("fusion", r"k_fusion"),
    ("optimizer", r"k_adam"),
    ("weight packing", r"k_pack"),
    ("runtime copies", r"copyBuffer|CopyDeviceToDevice"),
    ("ATen/other", r".*"),
]


def analyze(path, steps=1):
    with open(path) as f:
        rows = list(csv.DictReader(f))
    stage_ns = {name: 0.0 for name, _ in STAGES}
    stage_calls = {name: 0 for name, _ in STAGES}
    for r in rows:
        ns = float(r["TotalDurationNs"])
        calls = int(r["Calls"])
        for name, pat in STAGES:
            if re.search(pat, r["Name"]):
                stage_ns[name] += ns
                stage_calls[name] += calls
                break
    total = sum(stage_ns.values()) or 1.0
    print(f"{'stage':28s} {'ms/step':>9s} {'%':>6s} {'calls/step':>11s}")
    for name, _ in STAGES:
        if stage_ns[name] == 0:
            continue
        print(f"{name:28s} {stage_ns[name] / steps / 1e6:9.3f} "
              f"{100 * stage_ns[name] / total:6.2f} "
              f"{stage_calls[name] / steps:11.1f}")
    print(f"{'TOTAL (gpu busy)':28s} {total / steps / 1e6:9.3f}")
    return stage_ns


if __name__ == "__main__":
    analyze(sys.argv[1], int(sys.argv[2]) if len(sys.argv) > 2 else 1)
