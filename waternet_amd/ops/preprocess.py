"""GPU-native batch preprocess (white balance / gamma / CLAHE hist-eq).

gpu_transform_batch replaces the reference's per-image CPU transform loop
(data.py:81-90; the dataloader bottleneck per SURVEY §6) with 7 CDNA4
kernels over the whole uint8 batch on-device.
"""

import torch

from waternet_amd.ops import ext


def gpu_transform_batch(raw_u8: torch.Tensor):
    """raw_u8: (N,H,W,3) uint8 CUDA -> (wb, gc, he) uint8 same shape.
    Requires H, W divisible by 8 (CLAHE tile grid); callers fall back to the
    CPU transforms otherwise."""
    assert raw_u8.is_cuda and raw_u8.dtype == torch.uint8
    wb, gc, he = ext().preprocess_all(raw_u8.contiguous())
    return wb, gc, he
