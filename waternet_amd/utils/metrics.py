"""SSIM / PSNR metrics (native replacement for torchmetrics.functional).

The reference calls torchmetrics.functional.structural_similarity_index_measure
(11x11 gaussian window, sigma 1.5, k1=0.01, k2=0.03, data_range inferred when
not given) and peak_signal_noise_ratio(data_range=1) per minibatch
(train.py:9-12,67-68,141-144). torchmetrics is not installed here; these are
scratch implementations of the same definitions (Wang et al. 2004 SSIM with
gaussian weighting, uniform mean over the valid map).

On GPU with the native extension loaded, SSIM runs through the hand-written
HIP kernel (csrc/ssim.hip: per-window gaussian statistics + reduction);
otherwise the pure-torch composition below is used (CPU fallback / parity
reference for the kernel's unit test).
"""

import math
import os

import torch
import torch.nn.functional as F


def _gaussian_kernel2d(kernel_size: int, sigma: float, device, dtype):
    half = (kernel_size - 1) / 2.0
    coords = torch.arange(kernel_size, device=device, dtype=dtype) - half
    g = torch.exp(-(coords**2) / (2.0 * sigma * sigma))
    g = g / g.sum()
    k2d = torch.outer(g, g)
    return k2d


def structural_similarity_index_measure(
    preds: torch.Tensor,
    target: torch.Tensor,
    data_range: float = None,
    kernel_size: int = 11,
    sigma: float = 1.5,
    k1: float = 0.01,
    k2: float = 0.03,
) -> torch.Tensor:
    """Mean SSIM over the batch. data_range=None infers
    max(preds.max-preds.min, target.max-target.min) (torchmetrics behavior)."""
    if data_range is None:
        data_range = float(
            torch.maximum(
                preds.max() - preds.min(), target.max() - target.min()
            ).item()
        )

    if preds.is_cuda and os.environ.get("WATERNET_AMD_EAGER", "0") != "1":
        from waternet_amd.ops import native_available

        if native_available():
            from waternet_amd.ops.ssim import ssim_native

            return ssim_native(preds, target, data_range, kernel_size, sigma,
                               k1, k2)

    return _ssim_torch(preds, target, data_range, kernel_size, sigma, k1, k2)


def _ssim_torch(preds, target, data_range, kernel_size=11, sigma=1.5,
                k1=0.01, k2=0.03):
    preds = preds.float()
    target = target.float()
    n, c, h, w = preds.shape
    kern = _gaussian_kernel2d(kernel_size, sigma, preds.device, preds.dtype)
    kern = kern.expand(c, 1, kernel_size, kernel_size).contiguous()

    def filt(x):
        return F.conv2d(x, kern, groups=c)  # valid convolution

    mu_x = filt(preds)
    mu_y = filt(target)
    mu_xx = filt(preds * preds)
    mu_yy = filt(target * target)
    mu_xy = filt(preds * target)

    var_x = mu_xx - mu_x * mu_x
    var_y = mu_yy - mu_y * mu_y
    cov_xy = mu_xy - mu_x * mu_y

    c1 = (k1 * data_range) ** 2
    c2 = (k2 * data_range) ** 2
    num = (2 * mu_x * mu_y + c1) * (2 * cov_xy + c2)
    den = (mu_x * mu_x + mu_y * mu_y + c1) * (var_x + var_y + c2)
    return (num / den).mean()


def peak_signal_noise_ratio(
    preds: torch.Tensor, target: torch.Tensor, data_range: float = None
) -> torch.Tensor:
    """PSNR in dB over the whole batch (single global MSE, torchmetrics
    default reduction)."""
    if data_range is None:
        data_range = float((target.max() - target.min()).item())
    mse = torch.mean((preds.float() - target.float()) ** 2)
    return 10.0 * torch.log10(data_range**2 / mse)


def mse255(preds: torch.Tensor, target: torch.Tensor) -> torch.Tensor:
    """mean(square(255*(preds-target))) — the reference's MSE metric/loss
    scale (train.py:124)."""
    diff = 255.0 * (preds.float() - target.float())
    return torch.mean(diff * diff)
