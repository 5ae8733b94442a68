"""Tests of the data pipeline: bridge, augment, synthetic dataset, CLIs."""

import numpy as np
import torch

from waternet_amd.data.augment import PairedAugment
from waternet_amd.data.bridge import arr2ten, ten2arr
from waternet_amd.data.dataset import SyntheticUIEBDataset


def test_arr2ten_hwc():
    arr = np.arange(2 * 3 * 3, dtype=np.uint8).reshape(2, 3, 3)
    ten = arr2ten(arr)
    assert ten.shape == (3, 2, 3)
    assert torch.allclose(ten[0, 0, 0], torch.tensor(0.0))
    assert torch.allclose(ten[2, 1, 2], torch.tensor(17 / 255))


def test_arr2ten_batchdim():
    arr = np.zeros((4, 4, 3), dtype=np.uint8)
    assert arr2ten(arr, add_batch_dim=True).shape == (1, 3, 4, 4)
    batched = np.zeros((2, 4, 4, 3), dtype=np.uint8)
    assert arr2ten(batched).shape == (2, 3, 4, 4)


def test_ten2arr_roundtrip():
    arr = np.random.default_rng(0).integers(
        0, 256, size=(3, 8, 8, 3), dtype=np.uint8
    )
    back = ten2arr(arr2ten(arr))
    assert back.shape == arr.shape
    # /255 then *255 truncation: exact roundtrip
    assert np.array_equal(back, arr)


def test_ten2arr_clips():
    t = torch.tensor([[[[-0.5, 0.5], [1.5, 1.0]]]])
    arr = ten2arr(t)
    assert arr.min() == 0 and arr.max() == 255


def test_paired_augment_joint():
    rng = np.random.default_rng(0)
    aug = PairedAugment(rng=np.random.default_rng(42))
    img = rng.integers(0, 256, size=(16, 16, 3), dtype=np.uint8)
    # mask = image => transformed pair must stay identical
    for _ in range(20):
        a, b = aug(image=img, mask=img.copy())
        assert np.array_equal(a, b)


def test_paired_augment_actually_augments():
    img = np.arange(16 * 16 * 3, dtype=np.uint8).reshape(16, 16, 3)
    aug = PairedAugment(p_hflip=1.0, p_vflip=0.0, p_rot90=0.0)
    a, _ = aug(image=img, mask=img.copy())
    assert np.array_equal(a, img[:, ::-1])


def test_synthetic_dataset():
    ds = SyntheticUIEBDataset(n_images=4, im_height=32, im_width=32)
    item = ds[0]
    assert set(item.keys()) == {"raw", "wb", "gc", "he", "ref"}
    for v in item.values():
        assert v.shape == (3, 32, 32)
        assert v.dtype == torch.float32
        assert 0.0 <= v.min() and v.max() <= 1.0
    # determinism
    item2 = SyntheticUIEBDataset(n_images=4, im_height=32, im_width=32)[0]
    assert torch.equal(item["raw"], item2["raw"])


def test_synthetic_dataset_no_transforms():
    ds = SyntheticUIEBDataset(n_images=2, im_height=16, im_width=16,
                              run_transforms=False)
    assert set(ds[0].keys()) == {"raw", "ref"}


def test_hubconf_tuple():
    import hubconf

    preprocess, postprocess, model = hubconf.waternet(pretrained=False)
    rgb = np.random.default_rng(1).integers(
        0, 256, size=(32, 32, 3), dtype=np.uint8
    )
    rgb_ten, wb_ten, he_ten, gc_ten = preprocess(rgb)
    for t in (rgb_ten, wb_ten, he_ten, gc_ten):
        assert t.shape == (1, 3, 32, 32)
    model.eval()
    with torch.no_grad():
        out = model(rgb_ten, wb_ten, he_ten, gc_ten)
    arr = postprocess(out)
    assert arr.shape == (1, 32, 32, 3)
    assert arr.dtype == np.uint8


def test_synthetic_dataset_raw_mode():
    import torch

    from waternet_amd.data.dataset import SyntheticUIEBDataset

    ds = SyntheticUIEBDataset(n_images=3, im_height=16, im_width=16,
                              raw_mode=True)
    item = ds[0]
    assert set(item.keys()) == {"raw", "ref"}
    assert item["raw"].dtype == torch.uint8
    assert item["raw"].shape == (16, 16, 3)
    # collates to (B,H,W,3) uint8 for the fast engine's static buffers
    loader = torch.utils.data.DataLoader(ds, batch_size=2)
    batch = next(iter(loader))
    assert batch["raw"].shape == (2, 16, 16, 3)
    assert batch["raw"].dtype == torch.uint8


def test_uieb_dataset_raw_mode(tmp_path):
    import numpy as np
    import torch
    from PIL import Image

    from waternet_amd.data.dataset import UIEBDataset

    (tmp_path / "raw").mkdir()
    (tmp_path / "ref").mkdir()
    rng = np.random.default_rng(0)
    for i in range(2):
        for d in ("raw", "ref"):
            Image.fromarray(rng.integers(
                0, 256, size=(40, 40, 3), dtype=np.uint8
            )).save(tmp_path / d / f"{i}.png")
    ds = UIEBDataset(tmp_path / "raw", tmp_path / "ref", im_height=32,
                     im_width=32, raw_mode=True)
    item = ds[0]
    assert set(item.keys()) == {"raw", "ref"}
    assert item["raw"].dtype == torch.uint8
    assert item["raw"].shape == (32, 32, 3)


def test_train_engine_fast_requires_gpu():
    import pytest
    import torch

    if torch.cuda.is_available():
        pytest.skip("CPU-only check")
    import train as train_cli

    with pytest.raises(SystemExit):
        train_cli.main(["--engine", "fast", "--epochs", "1",
                        "--synthetic", "4", "--batch-size", "2",
                        "--height", "32", "--width", "32"])


def test_pad8_helper():
    import numpy as np

    from waternet_amd.engine.inferencer import pad8

    rng = np.random.default_rng(0)
    x = rng.integers(0, 256, size=(45, 67, 3), dtype=np.uint8)
    p, h, w = pad8(x)
    assert (h, w) == (45, 67)
    assert p.shape == (48, 72, 3)
    assert np.array_equal(p[:45, :67], x)
    # reflect padding: row 45 mirrors row 43 (reflect excludes the edge)
    assert np.array_equal(p[45], p[43])
    # already-aligned input passes through untouched
    y = rng.integers(0, 256, size=(64, 64, 3), dtype=np.uint8)
    p2, h2, w2 = pad8(y)
    assert p2 is y and (h2, w2) == (64, 64)


def test_hubconf_pretrained_offline_error():
    """pretrained=True (the reference default) must raise a CLEAR error
    when the download cannot proceed, not a bare network traceback."""
    import pytest

    import hubconf

    with pytest.raises(RuntimeError, match="checkpoint=|pretrained=False"):
        hubconf.waternet(pretrained=True, device="cpu")


def test_loader_with_workers_and_collation(tmp_path):
    """Multi-worker loading (the fast engine's default, train.py:222-230):
    PIL decode in worker processes, uint8 raw_mode batches collate to
    (B,H,W,3), ragged tail preserved, epoch re-iteration works with
    persistent workers."""
    from PIL import Image

    from waternet_amd.data.dataset import UIEBDataset

    rng = np.random.default_rng(5)
    (tmp_path / "raw-890").mkdir()
    (tmp_path / "reference-890").mkdir()
    for i in range(5):
        for d in ("raw-890", "reference-890"):
            Image.fromarray(rng.integers(
                0, 256, size=(40, 40, 3), dtype=np.uint8
            )).save(tmp_path / d / f"{i}.png")
    ds = UIEBDataset(tmp_path / "raw-890", tmp_path / "reference-890",
                     im_height=32, im_width=32, raw_mode=True)
    loader = torch.utils.data.DataLoader(ds, batch_size=2, num_workers=2,
                                         persistent_workers=True)
    for _ in range(2):  # two epochs through persistent workers
        sizes = [b["raw"].shape for b in loader]
        assert sizes == [torch.Size([2, 32, 32, 3])] * 2 + \
            [torch.Size([1, 32, 32, 3])]  # ragged tail
    del loader
