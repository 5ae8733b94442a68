"""Tests of the CPU-reference preprocess transforms (waternet_amd.data).

These pin down the exact semantics the HIP preprocess kernels must match
(SURVEY §2.3): white balance quantile-clip + stretch, gamma LUT, CLAHE.
"""

import numpy as np
import pytest

from waternet_amd.data.transforms import (
    clahe_u8,
    gamma_correction,
    histeq,
    lab2rgb_u8,
    rgb2lab_u8,
    transform,
    white_balance_transform,
)


def _rand_img(h=64, w=64, seed=0):
    rng = np.random.default_rng(seed)
    return rng.integers(0, 256, size=(h, w, 3), dtype=np.uint8)


def test_transform_order_and_types():
    """transform returns (wb, gc, he) in that order, uint8 HWC
    [data.py:81-90]."""
    img = _rand_img()
    wb, gc, he = transform(img)
    for t in (wb, gc, he):
        assert t.dtype == np.uint8
        assert t.shape == img.shape


def test_gamma_matches_reference_formula():
    img = _rand_img(seed=1)
    gc = gamma_correction(img)
    expected = np.clip(255 * np.power(img / 255, 0.7), 0, 255).astype(np.uint8)
    assert np.array_equal(gc, expected)


def test_gamma_monotone_and_endpoints():
    ramp = np.arange(256, dtype=np.uint8).reshape(16, 16)
    gc = gamma_correction(ramp)
    assert gc.flatten()[0] == 0
    assert gc.flatten()[255] == 255  # 255*(1)^0.7 = 255
    assert np.all(np.diff(gc.flatten().astype(int)) >= 0)


def test_white_balance_reference_semantics():
    """Replicate the reference's exact numpy math on a small random image
    (data.py:6-58) and compare elementwise."""
    img = _rand_img(h=32, w=32, seed=2)
    R = img[:, :, 0].astype(np.int64).sum()
    G = img[:, :, 1].astype(np.int64).sum()
    B = img[:, :, 2].astype(np.int64).sum()
    maxpix = max(R, G, B)
    ratio = np.array([maxpix / R, maxpix / G, maxpix / B])
    sat = 0.005 * ratio
    expected = np.zeros(img.shape)
    for ch in range(3):
        flat = img[:, :, ch].reshape(-1).astype(np.float64)
        lo, hi = np.quantile(flat, [sat[ch], 1 - sat[ch]])
        clipped = np.clip(flat, lo, hi)
        bottom, top = clipped.min(), clipped.max()
        expected[:, :, ch] = (
            (clipped - bottom) * 255 / (top - bottom)
        ).reshape(32, 32)
    got = white_balance_transform(img)
    assert np.array_equal(got, expected.astype(np.uint8))


def test_white_balance_stretches_to_full_range():
    img = (_rand_img(seed=3) // 2 + 64).astype(np.uint8)  # compressed range
    wb = white_balance_transform(img)
    for ch in range(3):
        assert wb[:, :, ch].min() == 0
        assert wb[:, :, ch].max() >= 254


def test_white_balance_does_not_mutate_input():
    """The reference grayscale path mutates its input in place
    (data.py:36,42-44) — documented latent bug, NOT replicated."""
    img = _rand_img(seed=4)[:, :, 0]
    orig = img.copy()
    white_balance_transform(img)
    assert np.array_equal(img, orig)


def test_lab_roundtrip_close():
    img = _rand_img(seed=5)
    back = lab2rgb_u8(rgb2lab_u8(img))
    # 8-bit LAB quantization loses some precision; stay within a few counts
    d = np.abs(back.astype(int) - img.astype(int))
    assert d.mean() < 2.0
    # out-of-gamut saturated colors clip harder through 8-bit LAB
    assert d.max() <= 25


def test_clahe_uniform_image_unchanged_mean():
    """CLAHE of a constant image maps the constant to ~its own value
    (clipped hist -> near-identity LUT by redistribution)."""
    img = np.full((64, 64), 128, dtype=np.uint8)
    out = clahe_u8(img)
    assert out.shape == img.shape
    # Analytic OpenCV result: tiles 8x8 (area 64), clip=1, excess 63
    # redistributed at step 4 -> cdf(128) = 34 -> round(34*255/64) = 135,
    # uniform across tiles so interpolation preserves it.
    assert np.all(out == 135)


def test_clahe_increases_contrast_of_lowcontrast_image():
    rng = np.random.default_rng(6)
    img = rng.integers(110, 146, size=(64, 64), dtype=np.uint8)
    out = clahe_u8(img)
    assert out.std() > img.std() * 1.5


def test_clahe_nondivisible_size():
    img = _rand_img(h=50, w=70, seed=7)[:, :, 0]
    out = clahe_u8(img)
    assert out.shape == (50, 70)


def test_histeq_preserves_shape_dtype():
    img = _rand_img(seed=8)
    he = histeq(img)
    assert he.shape == img.shape
    assert he.dtype == np.uint8


def test_histeq_only_touches_luminance():
    """a/b channels pass through: hue should be roughly preserved."""
    img = np.zeros((64, 64, 3), dtype=np.uint8)
    img[:, :, 0] = 150  # reddish image
    img[:, :, 1] = 60
    img[:, :, 2] = 60
    he = histeq(img)
    # red stays dominant
    assert he[:, :, 0].mean() > he[:, :, 1].mean()
    assert he[:, :, 0].mean() > he[:, :, 2].mean()


def test_white_balance_grayscale_path():
    """Reference data.py:32-44: grayscale input uses satLevels
    [0.001, 0.005] and must not mutate the input (the reference's in-place
    reshape-view mutation is a documented latent bug we do not replicate)."""
    import numpy as np

    from waternet_amd.data.transforms import white_balance_transform

    rng = np.random.default_rng(3)
    gray = rng.integers(20, 200, size=(32, 24), dtype=np.uint8)
    before = gray.copy()
    out = white_balance_transform(gray)
    assert out.shape == gray.shape and out.dtype == np.uint8
    assert np.array_equal(gray, before), "input mutated"
    # stretch reaches (close to) the full range
    assert out.max() >= 250 and out.min() <= 5


def test_wb_degenerate_channels():
    """All-black / single-channel images make the saturation ratio inf/nan
    (the reference crashes in np.quantile there, data.py:40); we fall back
    to a plain min/max stretch for the degenerate channel."""
    import numpy as np

    from waternet_amd.data.transforms import transform, white_balance_transform

    black = np.zeros((16, 16, 3), dtype=np.uint8)
    wb, gc, he = transform(black)
    assert wb.shape == black.shape and wb.dtype == np.uint8
    one = np.zeros((16, 16, 3), dtype=np.uint8)
    one[:, :, 1] = 200
    wbo = white_balance_transform(one)
    assert wbo.dtype == np.uint8
    assert wbo[:, :, 1].max() <= 255
