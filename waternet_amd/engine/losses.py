"""Training losses: VGG-perceptual + MSE composite.

Replicates train.py:110-127: ImageNet-normalize both images, run the VGG19
perceptual model on each, perceptual = mean(square(255*(f(x)-f(y)))),
mse = mean(square(255*(out-ref))), loss = 0.05*perceptual + mse.
"""

import torch

from waternet_amd.models.vgg import normalize_imagenet

PERCEPTUAL_WEIGHT = 0.05


def perceptual_loss(out, ref, vgg_model) -> torch.Tensor:
    fx = vgg_model(normalize_imagenet(out))
    with torch.no_grad():
        fy = vgg_model(normalize_imagenet(ref))
    d = 255.0 * (fx - fy)
    return torch.mean(d * d)


def mse_loss(out, ref) -> torch.Tensor:
    d = 255.0 * (out - ref)
    return torch.mean(d * d)


def composite_loss(out, ref, vgg_model):
    """Returns (loss, perceptual, mse)."""
    p = perceptual_loss(out, ref, vgg_model)
    m = mse_loss(out, ref)
    return PERCEPTUAL_WEIGHT * p + m, p, m
