"""Datasets: UIEB paired raw/reference images + synthetic stand-in.

UIEBDataset replicates the reference's dataset behavior
(training_utils.py:46-132) with PIL in place of OpenCV (not installed here):
pairs raw/ref files by name, resizes to (W, H) or to a multiple of 32 (VGG
requirement), applies joint flip/rot90 augmentation, runs the preprocess
transforms, and returns {raw, wb, gc, he, ref} CHW float tensors in [0,1].

SyntheticUIEBDataset produces deterministic random uint8 pairs of the same
shape for benchmarking / testing without the UIEB download (no network in
this environment).
"""

import os
from pathlib import Path
from typing import Optional

import numpy as np
import torch

from waternet_amd.data.augment import PairedAugment
from waternet_amd.data.bridge import arr2ten
from waternet_amd.data.transforms import transform as preprocess_transform


def _load_rgb(path) -> np.ndarray:
    from PIL import Image

    with Image.open(path) as im:
        return np.asarray(im.convert("RGB"))


def _resize_rgb(arr: np.ndarray, width: int, height: int) -> np.ndarray:
    from PIL import Image

    if arr.shape[0] == height and arr.shape[1] == width:
        return arr
    im = Image.fromarray(arr).resize((width, height), Image.BILINEAR)
    return np.asarray(im)


class UIEBDataset(torch.utils.data.Dataset):
    """Paired raw-890 / reference-890 dataset. [training_utils.py:46-132]"""

    def __init__(
        self,
        raw_dir,
        ref_dir,
        im_height: Optional[int] = None,
        im_width: Optional[int] = None,
        transform=None,
        raw_mode: bool = False,
    ):
        """raw_mode=True: skip the CPU preprocess transforms and return
        {raw, ref} uint8 HWC tensors — the fast GPU-preprocess train path
        runs wb/gamma/clahe on-device inside the step instead."""
        for d in (raw_dir, ref_dir):
            if not Path(d).is_dir():
                raise FileNotFoundError(
                    f"UIEB dataset directory not found: {d} — expected a "
                    "root containing raw-890/ and reference-890/ of paired "
                    ".png images (pass --data-root, or --synthetic N to "
                    "train without a dataset on disk)")
        raw_im_fns = sorted(
            p.name for p in Path(raw_dir).glob("*.png")
        )
        ref_im_fns = sorted(
            p.name for p in Path(ref_dir).glob("*.png")
        )
        if not raw_im_fns:
            raise FileNotFoundError(
                f"no .png images in {raw_dir} — expected the UIEB raw-890 "
                "set (or use --synthetic N)")
        assert set(raw_im_fns) == set(ref_im_fns), "raw/ref file mismatch"

        self.transform = transform if transform is not None else PairedAugment()
        self.raw_mode = raw_mode
        self.raw_dir = Path(raw_dir)
        self.ref_dir = Path(ref_dir)
        self.im_fns = raw_im_fns
        self.im_height = im_height
        self.im_width = im_width

    def __len__(self):
        return len(self.im_fns)

    def __getitem__(self, idx):
        raw_im = _load_rgb(self.raw_dir / self.im_fns[idx])
        ref_im = _load_rgb(self.ref_dir / self.im_fns[idx])

        if self.im_width is not None and self.im_height is not None:
            raw_im = _resize_rgb(raw_im, self.im_width, self.im_height)
            ref_im = _resize_rgb(ref_im, self.im_width, self.im_height)
        else:
            # Round spatial dims down to a multiple of 32 (VGG requirement,
            # training_utils.py:99-103)
            h, w = raw_im.shape[:2]
            vh, vw = (h // 32) * 32, (w // 32) * 32
            raw_im = _resize_rgb(raw_im, vw, vh)
            ref_im = _resize_rgb(ref_im, vw, vh)

        if self.transform is not None:
            raw_im, ref_im = self.transform(image=raw_im, mask=ref_im)

        if self.raw_mode:
            return {
                "raw": torch.from_numpy(np.ascontiguousarray(raw_im)),
                "ref": torch.from_numpy(np.ascontiguousarray(ref_im)),
            }

        wb, gc, he = preprocess_transform(raw_im)

        return {
            "raw": arr2ten(raw_im),
            "wb": arr2ten(wb),
            "gc": arr2ten(gc),
            "he": arr2ten(he),
            "ref": arr2ten(ref_im),
        }


class SyntheticUIEBDataset(torch.utils.data.Dataset):
    """Deterministic synthetic raw/ref uint8 pairs of UIEB shape.

    Used for benchmarking (no dataset download available): raw images are
    random uint8, ref images a smoothed variant. Preprocess transforms run
    exactly as for the real dataset (or can be skipped with
    precompute_transforms=False when the GPU preprocess path is used).
    """

    def __init__(self, n_images=800, im_height=112, im_width=112, seed=0,
                 run_transforms=True, raw_mode=False):
        self.n = n_images
        self.h = im_height
        self.w = im_width
        self.seed = seed
        self.run_transforms = run_transforms
        self.raw_mode = raw_mode
        # The dataset is a FIXED set of deterministic images (the synthetic
        # stand-in for files on disk); cache them after first generation so
        # later epochs read memory like a disk dataset reads page cache.
        # Unbounded only at small resolutions (800 x 2 x 112^2 x 3 = 60 MB).
        self._cache = {} if im_height * im_width <= 256 * 256 else None

    def __len__(self):
        return self.n

    def raw_uint8(self, idx):
        if self._cache is not None:
            hit = self._cache.get(idx)
            if hit is not None:
                return hit
        rng = np.random.default_rng(self.seed * 1_000_003 + idx)
        raw = rng.integers(0, 256, size=(self.h, self.w, 3), dtype=np.uint8)
        ref = rng.integers(0, 256, size=(self.h, self.w, 3), dtype=np.uint8)
        if self._cache is not None:
            self._cache[idx] = (raw, ref)
        return raw, ref

    def __getitem__(self, idx):
        raw_im, ref_im = self.raw_uint8(idx)
        if self.raw_mode:
            return {"raw": torch.from_numpy(raw_im),
                    "ref": torch.from_numpy(ref_im)}
        out = {"raw": arr2ten(raw_im), "ref": arr2ten(ref_im)}
        if self.run_transforms:
            wb, gc, he = preprocess_transform(raw_im)
            out.update({"wb": arr2ten(wb), "gc": arr2ten(gc), "he": arr2ten(he)})
        return out
