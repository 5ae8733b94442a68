"""uint8 HWC <-> float NCHW tensor bridge.

Replicates the reference's arr2ten/ten2arr semantics (training_utils.py:11-43;
the batch-dim-adding hub variant hubconf.py:8-34) in one place instead of
three duplicated copies.
"""

import numpy as np
import torch


def arr2ten(arr: np.ndarray, add_batch_dim: bool = False) -> torch.Tensor:
    """(N)HWC uint8 array -> float NCHW tensor in [0,1].

    add_batch_dim=True matches the inference/hub variant which unsqueezes a
    batch dim for HWC inputs (hubconf.py:17); False matches the training
    variant (training_utils.py:11-24).
    """
    ten = torch.from_numpy(np.ascontiguousarray(arr)) / 255
    if ten.ndim == 3:
        ten = ten.permute(2, 0, 1)
        if add_batch_dim:
            ten = ten.unsqueeze(0)
    elif ten.ndim == 4:
        ten = ten.permute(0, 3, 1, 2)
    return ten


def ten2arr(ten: torch.Tensor) -> np.ndarray:
    """float (N)CHW tensor -> uint8 (N)HWC array: clip to [0,1], *255,
    truncate to uint8. [training_utils.py:27-43]"""
    arr = ten.detach().cpu().float().numpy()
    arr = np.clip(arr, 0, 1)
    arr = (arr * 255).astype(np.uint8)
    if arr.ndim == 3:
        arr = np.transpose(arr, (1, 2, 0))
    elif arr.ndim == 4:
        arr = np.transpose(arr, (0, 2, 3, 1))
    return arr
