"""Band-wise inference correctness (CPU, eager path).

waternet_forward_banded splits large frames into horizontal bands with a
RECEPTIVE_HALO=13-row overlap so the conv activations stay L3-resident on
the GPU. The halo must cover WaterNet's receptive radius exactly — these
tests prove the band math (halo, crop, stitch, ragged last band) against
the eager fp32 model, where banding must reproduce the whole-frame output
(up to conv-algorithm accumulation-order noise; the GPU test in
test_gpu_engine.py asserts BIT-exactness on the native kernels, whose
per-pixel FMA order is band-independent)."""

import numpy as np
import pytest
import torch

from waternet_amd.engine.native import RECEPTIVE_HALO, auto_band_rows
from waternet_amd.models.waternet import WaterNet


def eager_banded(model, x, wb, ce, gc, band_rows, halo=RECEPTIVE_HALO):
    """NCHW eager replica of native.waternet_forward_banded's band loop."""
    h = x.shape[2]
    outs = []
    r0 = 0
    while r0 < h:
        r1 = min(r0 + band_rows, h)
        hs, he = max(r0 - halo, 0), min(r1 + halo, h)
        ob = model(x[:, :, hs:he], wb[:, :, hs:he],
                   ce[:, :, hs:he], gc[:, :, hs:he])
        outs.append(ob[:, :, r0 - hs: r0 - hs + (r1 - r0)])
        r0 = r1
    return torch.cat(outs, dim=2)


@pytest.mark.parametrize("h,w,band", [(96, 64, 32), (90, 48, 28),
                                      (64, 32, 64), (70, 40, 17)])
def test_banded_equals_whole(h, w, band):
    """Banding reproduces the whole-frame forward for even/odd heights,
    ragged last bands, and band >= H (single band)."""
    torch.manual_seed(0)
    model = WaterNet().eval()
    g = torch.Generator().manual_seed(1)
    ins = [torch.rand(1, 3, h, w, generator=g) for _ in range(4)]
    with torch.no_grad():
        whole = model(*ins)
        banded = eager_banded(model, *ins, band_rows=band)
    assert banded.shape == whole.shape
    torch.testing.assert_close(banded, whole, rtol=0, atol=1e-5)


def test_halo_too_short_diverges():
    """A halo inside the refiner's receptive radius (6) visibly corrupts
    band boundaries — guards against the constant silently shrinking below
    the true radius. (At halo 6..12 only the attenuated tail of the deep
    CMG path is cut: measured boundary error 1e-5..2e-7 at random init,
    reaching ~1.5e-8 = fp32 noise exactly at halo 13.)"""
    torch.manual_seed(0)
    model = WaterNet().eval()
    g = torch.Generator().manual_seed(1)
    ins = [torch.rand(1, 3, 96, 32, generator=g) for _ in range(4)]
    with torch.no_grad():
        whole = model(*ins)
        short = eager_banded(model, *ins, band_rows=32, halo=4)
    assert (short - whole).abs().max().item() > 1e-3


def test_auto_band_policy():
    """Whole-frame below the L3 working-set threshold, banded above."""
    assert auto_band_rows(112, 112) == 0       # flagship train shape
    assert auto_band_rows(512, 512) == 0       # fits the 256 MB L3
    rows_1080 = auto_band_rows(1088, 1920)
    assert rows_1080 > 0
    # the banded working set must actually fit the target
    live = (rows_1080 + 2 * RECEPTIVE_HALO) * 1920 * 256 * 2
    assert live <= 160 << 20
    assert rows_1080 >= 32


def test_banded_engine_requires_single_image():
    """The NHWC banded path asserts N == 1 (row slices of a batch are not
    contiguous)."""
    from waternet_amd.engine.native import waternet_forward_banded

    model = WaterNet().eval()
    t = torch.zeros(2, 32, 32, 16)
    with pytest.raises(AssertionError):
        waternet_forward_banded(model, t, t, t, t, band_rows=16)
