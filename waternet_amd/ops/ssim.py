"""Native HIP SSIM (metric path, no autograd) — the reference's
torchmetrics StructuralSimilarityIndexMeasure call site (train.py:44-47,
SURVEY.md §2.2 K21), same 11x11/sigma-1.5 gaussian-window semantics."""

import torch

from waternet_amd.ops import ext


def ssim_nhwc(preds, target, clog, data_range, k1=0.01, k2=0.03):
    """SSIM over NHWC bf16 tensors (clog logical channels) — the full-NHWC
    metric path (11x11 gaussian, sigma 1.5, torchmetrics semantics)."""
    a = preds.contiguous()
    b = target.contiguous()
    s = ext().ssim_sum_nhwc(a, b, clog, float(data_range), k1, k2)
    n, h, w, _ = a.shape
    count = n * clog * (h - 10) * (w - 10)
    return (s / count).to(torch.float32)


def ssim_native(preds, target, data_range, kernel_size=11, sigma=1.5,
                k1=0.01, k2=0.03):
    if kernel_size != 11 or abs(sigma - 1.5) > 1e-9:
        from waternet_amd.utils.metrics import _ssim_torch

        return _ssim_torch(preds, target, data_range, kernel_size, sigma, k1,
                           k2)
    a = preds.float().contiguous()
    b = target.float().contiguous()
    s = ext().ssim_sum(a, b, float(data_range), k1, k2)
    n, c, h, w = a.shape
    count = n * c * (h - kernel_size + 1) * (w - kernel_size + 1)
    return (s / count).to(torch.float32)
