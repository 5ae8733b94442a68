"""Joint image/target augmentations (native replacement for albumentations).

The reference uses A.Compose([HorizontalFlip(0.5), VerticalFlip(0.5),
RandomRotate90(0.5)]) applied jointly to raw/ref via image/mask
(training_utils.py:72-78,109-111). Re-implemented with numpy ops; each
op fires independently with p=0.5, RandomRotate90 picks k in {0,1,2,3}.
"""

from typing import Optional, Tuple

import numpy as np


class PairedAugment:
    def __init__(self, p_hflip=0.5, p_vflip=0.5, p_rot90=0.5,
                 rng: Optional[np.random.Generator] = None):
        self.p_hflip = p_hflip
        self.p_vflip = p_vflip
        self.p_rot90 = p_rot90
        self.rng = rng if rng is not None else np.random.default_rng()

    def __call__(
        self, image: np.ndarray, mask: np.ndarray
    ) -> Tuple[np.ndarray, np.ndarray]:
        if self.rng.random() < self.p_hflip:
            image = image[:, ::-1]
            mask = mask[:, ::-1]
        if self.rng.random() < self.p_vflip:
            image = image[::-1]
            mask = mask[::-1]
        if self.rng.random() < self.p_rot90:
            # albumentations RandomRotate90 rotates by a random k in 0..3
            k = int(self.rng.integers(0, 4))
            image = np.rot90(image, k)
            mask = np.rot90(mask, k)
        return np.ascontiguousarray(image), np.ascontiguousarray(mask)
