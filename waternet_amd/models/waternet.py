"""WaterNet gated-fusion enhancement network.

Architecture and state_dict schema replicate /root/reference/waternet/net.py:
  - ConfidenceMapGenerator (net.py:7-56): cat(x,wb,ce,gc) -> 8 convs
    (k7,k5,k3,k1,k7,k5,k3,k3; 12->128->128->128->64->64->64->64->3) with ReLU
    after 1-7 and Sigmoid after 8, split into three (N,1,H,W) confidence maps.
  - Refiner (net.py:59-80): cat(x,xbar) -> convs k7,k5,k3 (6->32->32->3),
    ReLU each.
  - WaterNet.forward(x, wb, ce, gc) = sum(refined_i * map_i) (net.py:99-108).

Parameters are stored as standard nn.Conv2d modules named exactly as the
reference so checkpoints interchange bit-for-bit:
  cmg.conv{1..8}.{weight,bias}, {wb,ce,gc}_refiner.conv{1..3}.{weight,bias}
(34 tensors, 1,090,668 params — train.py:308 schema).

Execution: on a ROCm GPU the forward runs through the hand-written CDNA4 HIP
engine (NHWC bf16 MFMA implicit-GEMM convolutions with fused bias+ReLU/
sigmoid epilogues and a fused gated-fusion kernel — waternet_amd.ops). On
CPU it runs a plain PyTorch reference composition with identical semantics.
"""

import torch
import torch.nn as nn
import torch.nn.functional as F


class ConfidenceMapGenerator(nn.Module):
    """Confidence-map branch: 8 same-pad convs over cat(x, wb, ce, gc)."""

    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(12, 128, kernel_size=7, padding="same")
        self.conv2 = nn.Conv2d(128, 128, kernel_size=5, padding="same")
        self.conv3 = nn.Conv2d(128, 128, kernel_size=3, padding="same")
        self.conv4 = nn.Conv2d(128, 64, kernel_size=1, padding="same")
        self.conv5 = nn.Conv2d(64, 64, kernel_size=7, padding="same")
        self.conv6 = nn.Conv2d(64, 64, kernel_size=5, padding="same")
        self.conv7 = nn.Conv2d(64, 64, kernel_size=3, padding="same")
        self.conv8 = nn.Conv2d(64, 3, kernel_size=3, padding="same")

    def forward(self, x, wb, ce, gc):
        out = torch.cat([x, wb, ce, gc], dim=1)
        out = F.relu(self.conv1(out))
        out = F.relu(self.conv2(out))
        out = F.relu(self.conv3(out))
        out = F.relu(self.conv4(out))
        out = F.relu(self.conv5(out))
        out = F.relu(self.conv6(out))
        out = F.relu(self.conv7(out))
        out = torch.sigmoid(self.conv8(out))
        return torch.split(out, [1, 1, 1], dim=1)


class Refiner(nn.Module):
    """Refinement branch: 3 same-pad convs over cat(x, xbar), all ReLU."""

    def __init__(self):
        super().__init__()
        self.conv1 = nn.Conv2d(6, 32, kernel_size=7, padding="same")
        self.conv2 = nn.Conv2d(32, 32, kernel_size=5, padding="same")
        self.conv3 = nn.Conv2d(32, 3, kernel_size=3, padding="same")

    def forward(self, x, xbar):
        out = torch.cat([x, xbar], dim=1)
        out = F.relu(self.conv1(out))
        out = F.relu(self.conv2(out))
        out = F.relu(self.conv3(out))
        return out


class WaterNet(nn.Module):
    """
    waternet = WaterNet()
    x = torch.randn(16, 3, 112, 112)
    waternet(x, x, x, x).shape  # torch.Size([16, 3, 112, 112])
    """

    def __init__(self):
        super().__init__()
        self.cmg = ConfidenceMapGenerator()
        self.wb_refiner = Refiner()
        self.ce_refiner = Refiner()
        self.gc_refiner = Refiner()

    def forward(self, x, wb, ce, gc):
        if _use_native(x):
            from waternet_amd.engine.native import waternet_forward_native

            return waternet_forward_native(self, x, wb, ce, gc)

        wb_cm, ce_cm, gc_cm = self.cmg(x, wb, ce, gc)
        refined_wb = self.wb_refiner(x, wb)
        refined_ce = self.ce_refiner(x, ce)
        refined_gc = self.gc_refiner(x, gc)
        return (
            refined_wb * wb_cm + refined_ce * ce_cm + refined_gc * gc_cm
        )


def _use_native(x: torch.Tensor) -> bool:
    """Native HIP engine policy: required on GPU (fail loudly if the
    extension is missing), never used on CPU. WATERNET_AMD_EAGER=1 forces
    the eager PyTorch path for debugging only."""
    import os

    if not x.is_cuda:
        return False
    if os.environ.get("WATERNET_AMD_EAGER", "0") == "1":
        return False
    from waternet_amd.ops import native_available, native_load_error

    if not native_available():
        raise RuntimeError(
            "waternet_amd: running on a GPU but the native HIP extension is "
            "not available — refusing to fall back to eager silently. Build "
            "it with `python -m waternet_amd.build` or set WATERNET_AMD_EAGER=1 "
            f"to force the eager path. Load error: {native_load_error()}"
        )
    return True
