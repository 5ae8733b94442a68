"""waternet_amd — MI355X-native underwater image enhancement framework.

A from-scratch AMD Instinct MI355X (gfx950 / CDNA4) implementation of the
capabilities of tnwei/waternet (WaterNet, IEEE TIP 2019): the gated-fusion
enhancement network, its three input transforms (white balance, gamma
correction, CLAHE histogram equalization), training with VGG-perceptual +
MSE loss, SSIM/PSNR metrics, data-parallel training over RCCL/xGMI, and
compatible train/inference/score entry points plus the torch.hub-style
(preprocess, postprocess, model) API.

Compute path: PyTorch-ROCm tensors + hand-written HIP/CDNA4 kernels
(MFMA implicit-GEMM convolutions, fused elementwise/reduction kernels,
GPU-native preprocess) + RCCL collectives for multi-GPU data parallelism.

Reference API contracts replicated (file:line cites into /root/reference):
  - WaterNet.forward(x, wb, ce, gc) -> (N,3,H,W)      [net.py:83-108]
  - state_dict schema: cmg.conv{1..8}.{weight,bias} +
    {wb,ce,gc}_refiner.conv{1..3}.{weight,bias}        [train.py:308]
  - transform(rgb) -> (wb, gc, he) uint8 HWC           [data.py:81-90]
  - hub tuple (preprocess, postprocess, model)         [hubconf.py:37-96]
"""

__version__ = "0.1.0"

from waternet_amd.models.waternet import WaterNet  # noqa: F401
from waternet_amd.data.transforms import (  # noqa: F401
    transform,
    white_balance_transform,
    gamma_correction,
    histeq,
)
from waternet_amd.data.bridge import arr2ten, ten2arr  # noqa: F401
