"""CPU tests of model structure, forward semantics and checkpoint schema."""

import torch
import pytest

from waternet_amd.models.waternet import WaterNet


EXPECTED_KEYS = (
    [f"cmg.conv{i}.{p}" for i in range(1, 9) for p in ("weight", "bias")]
    + [
        f"{b}_refiner.conv{i}.{p}"
        for b in ("wb", "ce", "gc")
        for i in range(1, 4)
        for p in ("weight", "bias")
    ]
)

EXPECTED_SHAPES = {
    "cmg.conv1.weight": (128, 12, 7, 7),
    "cmg.conv2.weight": (128, 128, 5, 5),
    "cmg.conv3.weight": (128, 128, 3, 3),
    "cmg.conv4.weight": (64, 128, 1, 1),
    "cmg.conv5.weight": (64, 64, 7, 7),
    "cmg.conv6.weight": (64, 64, 5, 5),
    "cmg.conv7.weight": (64, 64, 3, 3),
    "cmg.conv8.weight": (3, 64, 3, 3),
    "wb_refiner.conv1.weight": (32, 6, 7, 7),
    "wb_refiner.conv2.weight": (32, 32, 5, 5),
    "wb_refiner.conv3.weight": (3, 32, 3, 3),
}


def test_forward_shape():
    """net.py:84-90 docstring contract: (16,3,112,112) in -> same shape out."""
    torch.manual_seed(0)
    model = WaterNet()
    x = torch.randn(4, 3, 64, 64)
    out = model(x, x, x, x)
    assert out.shape == (4, 3, 64, 64)


def test_forward_resolution_agnostic():
    model = WaterNet()
    x = torch.randn(1, 3, 96, 128)
    assert model(x, x, x, x).shape == (1, 3, 96, 128)


def test_state_dict_schema():
    """Checkpoint contract: 34 tensors, exact reference key names
    (train.py:308)."""
    model = WaterNet()
    sd = model.state_dict()
    assert sorted(sd.keys()) == sorted(EXPECTED_KEYS)
    assert len(sd) == 34  # 8 cmg convs + 9 refiner convs, weight+bias each
    total = sum(v.numel() for v in sd.values())
    assert total == 1_090_668
    for k, shape in EXPECTED_SHAPES.items():
        assert tuple(sd[k].shape) == shape, k


def test_state_dict_roundtrip(tmp_path):
    torch.manual_seed(1)
    m1 = WaterNet()
    torch.save(m1.state_dict(), tmp_path / "w.pt")
    m2 = WaterNet()
    m2.load_state_dict(torch.load(tmp_path / "w.pt", map_location="cpu"))
    x = torch.randn(1, 3, 32, 32)
    assert torch.equal(m1(x, x, x, x), m2(x, x, x, x))


def test_gated_fusion_semantics():
    """Output = sum of refined_i * map_i with maps from sigmoid in (0,1)."""
    torch.manual_seed(2)
    model = WaterNet()
    x = torch.rand(2, 3, 32, 32)
    wb, ce, gc = torch.rand_like(x), torch.rand_like(x), torch.rand_like(x)
    wb_cm, ce_cm, gc_cm = model.cmg(x, wb, ce, gc)
    for cm in (wb_cm, ce_cm, gc_cm):
        assert cm.shape == (2, 1, 32, 32)
        assert cm.min() > 0 and cm.max() < 1
    manual = (
        model.wb_refiner(x, wb) * wb_cm
        + model.ce_refiner(x, ce) * ce_cm
        + model.gc_refiner(x, gc) * gc_cm
    )
    assert torch.allclose(model(x, wb, ce, gc), manual, atol=1e-6)


def test_refiner_relu_output_nonnegative():
    model = WaterNet()
    x = torch.randn(1, 3, 32, 32)
    assert model.wb_refiner(x, x).min() >= 0


def test_backward_produces_grads():
    model = WaterNet()
    x = torch.rand(1, 3, 32, 32)
    out = model(x, x, x, x)
    out.mean().backward()
    grads = [p.grad for p in model.parameters()]
    assert all(g is not None for g in grads)
    assert all(torch.isfinite(g).all() for g in grads)


def test_torch_hub_local_load():
    """torch.hub.load(source='local') end-to-end through hubconf.py."""
    import pathlib

    import torch

    repo = str(pathlib.Path(__file__).resolve().parent.parent)
    pre, post, model = torch.hub.load(
        repo, "waternet", source="local", pretrained=False, device="cpu"
    )
    import numpy as np

    rng = np.random.default_rng(0)
    rgb = rng.integers(0, 256, size=(32, 32, 3), dtype=np.uint8)
    rgb_t, wb_t, he_t, gc_t = pre(rgb)
    with torch.no_grad():
        out = model(rgb_t, wb_t, he_t, gc_t)
    arr = post(out)
    assert arr.shape == (1, 32, 32, 3) and arr.dtype == np.uint8


def test_vgg_load_torchvision_schema():
    """load_torchvision_state_dict against a state_dict with torchvision's
    EXACT vgg19 key set (features.{0,2,5,7,10,12,14,16,19,21,23,25,28,30,
    32,34}.{weight,bias} + classifier.{0,3,6}.*) — proves the loader maps
    the real schema."""
    import torch

    from waternet_amd.models.vgg import PerceptualModel

    conv_idx = [0, 2, 5, 7, 10, 12, 14, 16, 19, 21, 23, 25, 28, 30, 32, 34]
    chans = [(64, 3), (64, 64), (128, 64), (128, 128),
             (256, 128), (256, 256), (256, 256), (256, 256),
             (512, 256), (512, 512), (512, 512), (512, 512),
             (512, 512), (512, 512), (512, 512), (512, 512)]
    g = torch.Generator().manual_seed(77)
    sd = {}
    for idx, (co, ci) in zip(conv_idx, chans):
        sd[f"features.{idx}.weight"] = torch.randn(co, ci, 3, 3, generator=g)
        sd[f"features.{idx}.bias"] = torch.randn(co, generator=g)
    # classifier keys present in a full torchvision dict; must be ignored
    sd["classifier.0.weight"] = torch.randn(1, 1)
    sd["classifier.0.bias"] = torch.randn(1)
    sd["classifier.3.weight"] = torch.randn(1, 1)
    sd["classifier.3.bias"] = torch.randn(1)
    sd["classifier.6.weight"] = torch.randn(1, 1)
    sd["classifier.6.bias"] = torch.randn(1)

    m = PerceptualModel(seed=0)
    m.load_torchvision_state_dict(sd)
    assert torch.equal(m.model[0].weight, sd["features.0.weight"])
    assert torch.equal(m.model[34].weight, sd["features.34.weight"])
    assert torch.equal(m.model[19].bias, sd["features.19.bias"])
    # and the features-only subset form works too
    m2 = PerceptualModel(seed=0)
    feats_only = {k[len("features."):]: v for k, v in sd.items()
                  if k.startswith("features.")}
    m2.load_torchvision_state_dict(feats_only)
    assert torch.equal(m2.model[34].weight, sd["features.34.weight"])
