"""CPU reference implementations of WaterNet's three input transforms.

Semantics replicate /root/reference/waternet/data.py (white balance
data.py:6-58, gamma data.py:61-65, CLAHE hist-eq data.py:68-78) but are
re-implemented from scratch in pure NumPy: this environment has no OpenCV,
so the LAB color conversion and CLAHE are implemented here natively with
OpenCV-equivalent algorithms (per-tile 256-bin histograms, integer clip +
redistribute, bilinear LUT interpolation). The HIP GPU kernels in
waternet_amd/csrc/preprocess.hip are validated against THIS module.

Notes on fidelity:
  - White balance follows SimplestColorBalance.m semantics exactly as the
    reference does (per-channel quantile clip at 0.005*maxsum/chansum both
    ends + min-max stretch). The reference's in-place input mutation on the
    grayscale path (data.py:36,42-44) is a latent bug and is NOT replicated.
  - CLAHE follows OpenCV's algorithm (clip limit max(int(0.1*tileArea/256),1),
    excess redistribution with residual stepping, round-half-even LUT).
    The reference README itself documents its CLAHE is not bit-equivalent to
    the original MATLAB adapthisteq (README.md:138); likewise this float-math
    LAB conversion is equivalent-but-not-bitwise vs OpenCV's fixed-point LUTs.
"""

from typing import Tuple

import numpy as np

# --------------------------------------------------------------------------
# White balance
# --------------------------------------------------------------------------


def white_balance_transform(im_rgb: np.ndarray) -> np.ndarray:
    """Simplest color balance. HWC uint8 (or HW grayscale uint8) -> uint8.

    Per channel: clip at the [satLo, 1-satHi] quantiles (linear-interpolation
    quantiles, as np.quantile computes them), then min-max stretch to [0,255].
    satLo = satHi = 0.005 * (max_channel_sum / channel_sum) for RGB;
    [0.001, 0.005] for grayscale.
    """
    if im_rgb.ndim == 3:
        h, w, p = im_rgb.shape
        chan_sums = np.array(
            [im_rgb[:, :, i].astype(np.int64).sum() for i in range(p)],
            dtype=np.float64,
        )
        with np.errstate(divide="ignore", invalid="ignore"):
            ratio = chan_sums.max() / chan_sums  # inf/nan handled below
        sat_lo = 0.005 * ratio
        sat_hi = 0.005 * ratio
        flat = im_rgb.reshape(h * w, p).T
    else:
        h, w = im_rgb.shape
        p = 1
        sat_lo = np.array([0.001])
        sat_hi = np.array([0.005])
        flat = im_rgb.reshape(1, h * w)

    # uint8 data: quantiles are order statistics over a 256-bin histogram
    # (O(n) bincount instead of np.quantile's O(n log n) sort) and the
    # clip+stretch is a 256-entry float64 LUT gather — BIT-identical to
    # the per-pixel float64 formula (verified against the direct
    # np.quantile composition by test_white_balance_reference_semantics
    # and the hypothesis fuzz).
    n = flat.shape[1]
    out = np.empty((p, n), dtype=np.float64)
    vals = np.arange(256, dtype=np.float64)
    for ch in range(p):
        lo_q, hi_q = sat_lo[ch], 1.0 - sat_hi[ch]
        # Degenerate channels (a zero channel sum makes the sat ratio
        # inf/nan — the reference crashes in np.quantile here, data.py:40):
        # fall back to the un-saturated quantiles, i.e. plain min/max
        # stretch for that channel.
        if not (np.isfinite(lo_q) and 0.0 <= lo_q <= 0.5):
            lo_q, hi_q = 0.0, 1.0
        hist = np.bincount(flat[ch], minlength=256)
        cum = np.cumsum(hist)
        lo_v = _quantile_from_hist(cum, n, lo_q)
        hi_v = _quantile_from_hist(cum, n, hi_q)
        nz = np.nonzero(hist)[0]
        vmin, vmax = float(nz[0]), float(nz[-1])
        bottom = min(max(vmin, lo_v), hi_v)  # clip is monotone
        top = min(max(vmax, lo_v), hi_v)
        scale = 255.0 / (top - bottom) if top > bottom else 0.0
        lut = (np.clip(vals, lo_v, hi_v) - bottom) * scale
        out[ch] = lut[flat[ch]]

    if im_rgb.ndim == 3:
        return out.T.reshape(h, w, p).astype(np.uint8)
    return out.reshape(h, w).astype(np.uint8)


def _quantile_from_hist(cum: np.ndarray, n: int, q: float) -> float:
    """np.quantile('linear') over uint8 data from its cumulative histogram:
    pos = q*(n-1); order statistics a[floor]/a[ceil] via cumulative counts;
    numpy's _lerp semantics (the t>=0.5 branch) replicated exactly."""
    pos = q * (n - 1)
    lo_i = int(np.floor(pos))
    frac = pos - lo_i
    hi_i = lo_i + 1 if frac > 0.0 else lo_i
    v_lo = float(np.searchsorted(cum, lo_i, side="right"))
    v_hi = float(np.searchsorted(cum, hi_i, side="right"))
    if frac >= 0.5:  # numpy _lerp's accuracy branch
        return v_hi - (v_hi - v_lo) * (1.0 - frac)
    return v_lo + (v_hi - v_lo) * frac


# --------------------------------------------------------------------------
# Gamma correction
# --------------------------------------------------------------------------

GAMMA = 0.7


_GAMMA_LUT = None


def gamma_correction(im: np.ndarray) -> np.ndarray:
    """gc = clip(255 * (im/255)^0.7, 0, 255) truncated to uint8.

    uint8 input -> the map has only 256 possible values: computed once in
    float64 (bit-identical to the per-pixel formula) and applied as a LUT
    gather. The GPU kernel uses this same host-computed table."""
    global _GAMMA_LUT
    if _GAMMA_LUT is None:
        v = np.power(np.arange(256, dtype=np.float64) / 255.0, GAMMA)
        _GAMMA_LUT = np.clip(255.0 * v, 0, 255).astype(np.uint8)
    return _GAMMA_LUT[im]


# --------------------------------------------------------------------------
# RGB <-> LAB (D65, sRGB gamma, 8-bit scaling as OpenCV does)
# --------------------------------------------------------------------------

# sRGB -> XYZ (D65) matrix, rows produce X, Y, Z from linear RGB
_RGB2XYZ = np.array(
    [
        [0.412453, 0.357580, 0.180423],
        [0.212671, 0.715160, 0.072169],
        [0.019334, 0.119193, 0.950227],
    ]
)
_XYZ2RGB = np.linalg.inv(_RGB2XYZ)
# D65 white point
_WHITE = np.array([0.950456, 1.0, 1.088754])


def _srgb_linearize(s: np.ndarray) -> np.ndarray:
    """Inverse sRGB gamma, input in [0,1]."""
    return np.where(s <= 0.04045, s / 12.92, ((s + 0.055) / 1.055) ** 2.4)


def _srgb_delinearize(lin: np.ndarray) -> np.ndarray:
    lin = np.clip(lin, 0.0, 1.0)
    return np.where(
        lin <= 0.0031308, lin * 12.92, 1.055 * np.power(lin, 1.0 / 2.4) - 0.055
    )


def _lab_f(t: np.ndarray) -> np.ndarray:
    d = 6.0 / 29.0
    return np.where(t > d**3, np.cbrt(t), t / (3 * d * d) + 4.0 / 29.0)


def _lab_finv(ft: np.ndarray) -> np.ndarray:
    d = 6.0 / 29.0
    return np.where(ft > d, ft**3, 3 * d * d * (ft - 4.0 / 29.0))


_SRGB_LIN_LUT = None


def rgb2lab_u8(rgb: np.ndarray) -> np.ndarray:
    """HWC uint8 RGB -> HWC uint8 LAB with OpenCV 8-bit scaling
    (L*255/100, a+128, b+128). The sRGB linearization of a uint8 input
    has 256 possible values — applied as a float64 LUT gather,
    bit-identical to calling _srgb_linearize per pixel."""
    global _SRGB_LIN_LUT
    if _SRGB_LIN_LUT is None:
        _SRGB_LIN_LUT = _srgb_linearize(
            np.arange(256, dtype=np.float64) / 255.0)
    lin = _SRGB_LIN_LUT[rgb]
    xyz = lin @ _RGB2XYZ.T
    fxyz = _lab_f(xyz / _WHITE)
    L = 116.0 * fxyz[..., 1] - 16.0
    a = 500.0 * (fxyz[..., 0] - fxyz[..., 1])
    b = 200.0 * (fxyz[..., 1] - fxyz[..., 2])
    lab = np.stack([L * 255.0 / 100.0, a + 128.0, b + 128.0], axis=-1)
    return np.clip(np.rint(lab), 0, 255).astype(np.uint8)


def lab2rgb_u8(lab: np.ndarray) -> np.ndarray:
    """HWC uint8 LAB (OpenCV 8-bit scaling) -> HWC uint8 RGB."""
    L = lab[..., 0].astype(np.float64) * 100.0 / 255.0
    a = lab[..., 1].astype(np.float64) - 128.0
    b = lab[..., 2].astype(np.float64) - 128.0
    fy = (L + 16.0) / 116.0
    fx = fy + a / 500.0
    fz = fy - b / 200.0
    xyz = np.stack([_lab_finv(fx), _lab_finv(fy), _lab_finv(fz)], axis=-1) * _WHITE
    lin = xyz @ _XYZ2RGB.T
    srgb = _srgb_delinearize(lin)
    return np.clip(np.rint(srgb * 255.0), 0, 255).astype(np.uint8)


# --------------------------------------------------------------------------
# CLAHE (OpenCV-equivalent algorithm)
# --------------------------------------------------------------------------


def clahe_u8(
    src: np.ndarray, clip_limit: float = 0.1, tile_grid: Tuple[int, int] = (8, 8)
) -> np.ndarray:
    """Contrast-limited adaptive histogram equalization on a single-channel
    uint8 image, following OpenCV's algorithm:

      1. Pad the image (reflect-101) so H, W divide evenly by the tile grid.
      2. Per tile: 256-bin histogram; integer clip limit
         max(int(clip_limit * tileArea / 256), 1); clip and redistribute the
         excess (batch add + residual stepping); LUT[i] = round_half_even(
         cdf[i] * 255 / tileArea).
      3. Per pixel: bilinear interpolation between the 4 surrounding tile LUTs.
    """
    h, w = src.shape
    ty_n, tx_n = tile_grid
    # Pad to a multiple of the grid (reflect-101, like cv2.copyMakeBorder)
    ph = (ty_n - h % ty_n) % ty_n
    pw = (tx_n - w % tx_n) % tx_n
    img = np.pad(src, ((0, ph), (0, pw)), mode="reflect") if (ph or pw) else src
    H, W = img.shape
    th, tw = H // ty_n, W // tx_n
    tile_area = th * tw

    clip = max(int(clip_limit * tile_area / 256.0), 1) if clip_limit > 0 else 0

    # Per-tile histograms: (ty_n, tx_n, 256)
    tiles = img.reshape(ty_n, th, tx_n, tw).transpose(0, 2, 1, 3).reshape(
        ty_n, tx_n, tile_area
    )
    hist = np.zeros((ty_n, tx_n, 256), dtype=np.int64)
    for ti in range(ty_n):
        for tj in range(tx_n):
            hist[ti, tj] = np.bincount(tiles[ti, tj], minlength=256)

    if clip > 0:
        for ti in range(ty_n):
            for tj in range(tx_n):
                hh = hist[ti, tj]
                excess = int(np.maximum(hh - clip, 0).sum())
                if excess > 0:
                    np.minimum(hh, clip, out=hh)
                    batch = excess // 256
                    residual = excess - batch * 256
                    hh += batch
                    if residual:
                        step = max(256 // residual, 1)
                        idx = np.arange(0, 256, step)[:residual]
                        hh[idx] += 1

    lut_scale = 255.0 / tile_area
    cdf = np.cumsum(hist, axis=-1)
    # cv::saturate_cast<uchar>(float) rounds half-to-even (cvRound)
    luts = np.clip(np.rint(cdf * lut_scale), 0, 255).astype(np.uint8)

    # Bilinear interpolation of per-tile LUT outputs
    ys = np.arange(H, dtype=np.float64)
    xs = np.arange(W, dtype=np.float64)
    tyf = ys / th - 0.5
    txf = xs / tw - 0.5
    ty1 = np.floor(tyf).astype(np.int64)
    tx1 = np.floor(txf).astype(np.int64)
    ya = (tyf - ty1)[:, None]
    xa = (txf - tx1)[None, :]
    ty2 = np.clip(ty1 + 1, 0, ty_n - 1)
    tx2 = np.clip(tx1 + 1, 0, tx_n - 1)
    ty1 = np.clip(ty1, 0, ty_n - 1)
    tx1 = np.clip(tx1, 0, tx_n - 1)

    v = img  # (H, W) uint8 values index the LUTs
    # Gather LUT outputs for the 4 neighbor tiles
    l11 = luts[ty1[:, None], tx1[None, :], v].astype(np.float64)
    l12 = luts[ty1[:, None], tx2[None, :], v].astype(np.float64)
    l21 = luts[ty2[:, None], tx1[None, :], v].astype(np.float64)
    l22 = luts[ty2[:, None], tx2[None, :], v].astype(np.float64)
    res = (
        l11 * (1 - xa) * (1 - ya)
        + l12 * xa * (1 - ya)
        + l21 * (1 - xa) * ya
        + l22 * xa * ya
    )
    out = np.clip(np.rint(res), 0, 255).astype(np.uint8)
    return out[:h, :w]


def histeq(im_rgb: np.ndarray) -> np.ndarray:
    """RGB -> LAB, CLAHE(clip 0.1, 8x8 tiles) on L, -> RGB. [data.py:68-78]"""
    lab = rgb2lab_u8(im_rgb)
    lab[:, :, 0] = clahe_u8(lab[:, :, 0], clip_limit=0.1, tile_grid=(8, 8))
    return lab2rgb_u8(lab)


# --------------------------------------------------------------------------
# Combined transform — the order (wb, gc, he) is the API contract
# --------------------------------------------------------------------------


def transform(rgb: np.ndarray) -> Tuple[np.ndarray, np.ndarray, np.ndarray]:
    """transform(rgb) -> (wb, gc, he), each HWC uint8. [data.py:81-90]"""
    wb = white_balance_transform(rgb)
    gc = gamma_correction(rgb)
    he = histeq(rgb)
    return wb, gc, he
