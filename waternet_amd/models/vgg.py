"""VGG19-features perceptual model (native replacement for torchvision).

The reference builds torchvision.models.vgg19(pretrained=True).features[:-1]
(train.py:254-263): all 16 3x3 convs + ReLUs and the first 4 maxpools of
VGG19 configuration E, dropping the final maxpool, output (N,512,H/16,W/16).

torchvision is not installed here and there is no network for pretrained
weights, so this module defines the same architecture natively with the
same child naming as torchvision (`features.<idx>`), so a torchvision VGG19
state_dict (features.* subset) loads directly via load_torchvision_state_dict.
Without one, ImageNet-pretrained behavior is unavailable; benchmarks use
seeded random init (BASELINE: synthetic data / random-init weights).

On GPU the forward runs through the HIP conv/maxpool kernels (3x3 MFMA
implicit-GEMM + 2x2 maxpool); this is the FLOPs-dominant path of training
(SURVEY §2.2 K17).
"""

import torch
import torch.nn as nn
import torch.nn.functional as F

# VGG19 cfg "E"; numbers are conv output channels, "M" is 2x2 maxpool.
_VGG19_CFG = [64, 64, "M", 128, 128, "M", 256, 256, 256, 256, "M",
              512, 512, 512, 512, "M", 512, 512, 512, 512, "M"]

IMAGENET_MEAN = (0.485, 0.456, 0.406)
IMAGENET_STD = (0.229, 0.224, 0.225)


def _make_features(drop_last_pool: bool = True) -> nn.Sequential:
    layers = []
    in_ch = 3
    for v in _VGG19_CFG:
        if v == "M":
            layers.append(nn.MaxPool2d(kernel_size=2, stride=2))
        else:
            layers.append(nn.Conv2d(in_ch, v, kernel_size=3, padding=1))
            layers.append(nn.ReLU(inplace=True))
            in_ch = v
    if drop_last_pool:
        layers = layers[:-1]
    return nn.Sequential(*layers)


class PerceptualModel(nn.Module):
    """VGG19 features[:-1]; same output as the reference's PerceptualModel
    (train.py:254-263). Weights get gradients but are never optimized
    (the reference optimizer only holds WaterNet params — train.py:250);
    we freeze them explicitly, which skips their wgrad for speed while the
    input gradient still flows (SURVEY §2.2 backward note)."""

    def __init__(self, seed: int = 0):
        super().__init__()
        g = torch.Generator().manual_seed(seed)
        self.model = _make_features(drop_last_pool=True)
        # Deterministic random init (no pretrained weights offline)
        for m in self.model.modules():
            if isinstance(m, nn.Conv2d):
                nn.init.kaiming_normal_(m.weight, generator=g)
                nn.init.zeros_(m.bias)
        for p in self.parameters():
            p.requires_grad_(False)

    def forward(self, x):
        if _use_native(x):
            from waternet_amd.engine.native import vgg_forward_native

            return vgg_forward_native(self, x)
        return self.model(x)

    def load_torchvision_state_dict(self, sd):
        """Load a torchvision vgg19 state_dict (accepts the full model dict
        with `features.*` keys, or just the features subset)."""
        feats = {}
        for k, v in sd.items():
            if k.startswith("features."):
                feats[k[len("features."):]] = v
            elif k.split(".")[0].isdigit():
                feats[k] = v
        missing, unexpected = self.model.load_state_dict(feats, strict=False)
        # Only keys past the dropped final maxpool may be unexpected
        if missing:
            raise RuntimeError(f"VGG19 load missing keys: {missing}")
        return unexpected


_NORM_CACHE = {}


def normalize_imagenet(x: torch.Tensor) -> torch.Tensor:
    """TF.normalize(x, ImageNet mean/std) equivalent (train.py:111-116).

    The mean/std constants are cached per (device, dtype): building them
    with new_tensor() is a pageable H2D copy, which is illegal inside
    hipGraph capture (it silently broke whole-step capture)."""
    key = (x.device, x.dtype)
    cached = _NORM_CACHE.get(key)
    if cached is None:
        mean = x.new_tensor(IMAGENET_MEAN).view(1, 3, 1, 1)
        std = x.new_tensor(IMAGENET_STD).view(1, 3, 1, 1)
        cached = _NORM_CACHE[key] = (mean, std)
    mean, std = cached
    return (x - mean) / std


def _use_native(x: torch.Tensor) -> bool:
    import os

    if not x.is_cuda or os.environ.get("WATERNET_AMD_EAGER", "0") == "1":
        return False
    from waternet_amd.ops import native_available

    return native_available()
