"""Small-unit CPU tests: the pad8 inference helper, the --show-split
composite, the bare (unbatched) tensor bridges, and the torchrun env
plumbing of the parallel layer."""

import numpy as np
import pytest
import torch

from waternet_amd.data.bridge import arr2ten, ten2arr
from waternet_amd.engine.inferencer import pad8
from waternet_amd.parallel import DistEnv, distributed_env


def test_pad8_identity_when_divisible():
    im = np.zeros((16, 24, 3), dtype=np.uint8)
    out, h, w = pad8(im)
    assert out is im and (h, w) == (16, 24)


def test_pad8_reflect_and_crop_roundtrip():
    rng = np.random.default_rng(0)
    im = rng.integers(0, 256, size=(13, 21, 3), dtype=np.uint8)
    out, h, w = pad8(im)
    assert (h, w) == (13, 21)
    assert out.shape == (16, 24, 3)  # next multiples of 8
    np.testing.assert_array_equal(out[:13, :21], im)  # crop restores
    # reflect semantics on the first padded row: mirror of row h-2
    np.testing.assert_array_equal(out[13, :21], im[11])


def test_compose_split_halves():
    """Left half = original, right half = enhanced (divider/text aside) —
    reference inference.py:202-233."""
    from inference import compose_split

    before = np.zeros((40, 60, 3), dtype=np.uint8)
    after = np.full((40, 60, 3), 200, dtype=np.uint8)
    comp = compose_split(before, after)
    assert comp.shape == (40, 60, 3)
    # below the text region, away from the divider line
    assert (comp[30:, :28] == 0).all()
    assert (comp[30:, 33:] == 200).all()


def test_bridge_unbatched_paths():
    rng = np.random.default_rng(1)
    im = rng.integers(0, 256, size=(9, 7, 3), dtype=np.uint8)
    ten = arr2ten(im)  # no batch dim: (3, 9, 7)
    assert ten.shape == (3, 9, 7)
    back = ten2arr(ten)
    assert back.shape == (9, 7, 3)
    np.testing.assert_array_equal(back, im)
    # ten2arr truncates (not rounds): 0.9999 * 255 = 254.97 -> 254
    t = torch.full((3, 2, 2), 0.9999)
    assert ten2arr(t).max() == 254


def test_distributed_env_parsing(monkeypatch):
    monkeypatch.setenv("RANK", "3")
    monkeypatch.setenv("LOCAL_RANK", "1")
    monkeypatch.setenv("WORLD_SIZE", "8")
    env = distributed_env()
    assert (env.rank, env.local_rank, env.world_size) == (3, 1, 8)
    assert not env.initialized


def test_average_metrics_world1_passthrough():
    env = DistEnv(rank=0, local_rank=0, world_size=1)
    m = {"a": 1.5, "b": -2.0}
    assert env.average_metrics(m) == m
    env.barrier()  # no-op without a process group


def test_convspec_pack_protocol(monkeypatch):
    """ConvSpec's dirty/version protocol (round-1 advisor subject): packs
    are cached by (weight._version, data_ptr), refreshed on mutation or
    mark_dirty, and the buffers are REWRITTEN IN PLACE — the batched
    pack_all descriptor bakes their device pointers, so the objects must
    never be reallocated."""
    import waternet_amd.ops as ops
    from waternet_amd.ops.conv import ACT_RELU, ConvSpec

    calls = {"fwd": 0, "dgrad": 0}

    class FakeExt:
        @staticmethod
        def pack_weight_fwd(w, kp, cp):
            calls["fwd"] += 1
            return torch.full((kp * 9 * cp,), float(calls["fwd"]))

        @staticmethod
        def pack_weight_dgrad(w, kp, cp):
            calls["dgrad"] += 1
            return torch.full((cp * 9 * kp,), float(calls["dgrad"]))

    monkeypatch.setattr(ops, "_ext", FakeExt())
    monkeypatch.setattr(ops, "_tried", True)
    mod = torch.nn.Conv2d(3, 5, 3, padding="same")
    spec = ConvSpec(mod, ACT_RELU)
    assert (spec.Kp, spec.Cp) == (16, 16)  # pow2 >= 16 channel padding

    wp1 = spec.packed_fwd()
    ptr = wp1.data_ptr()
    assert calls["fwd"] == 1
    spec.packed_fwd()
    assert calls["fwd"] == 1  # cached: same version -> no repack

    with torch.no_grad():
        mod.weight += 1.0  # bumps weight._version
    wp2 = spec.packed_fwd()
    assert calls["fwd"] == 2  # mutation detected
    assert wp2.data_ptr() == ptr  # rewritten IN PLACE, never reallocated

    spec.mark_dirty()  # FusedAdam's in-place master update (no version bump)
    spec.packed_fwd()
    assert calls["fwd"] == 3
    assert spec.packed_fwd().data_ptr() == ptr


def test_pack_descriptor_protocol(monkeypatch):
    """build_pack_descriptor (advisor r1 medium-2 subject): forces packs
    to exist FIRST, bakes their stable pointers into int64 rows
    [wptr, wp, wd, K, C, ks, ks, Kp, Cp]; mark_specs_packed records the
    weight versions so the lazy per-layer refresh skips after a batched
    pack_all rewrote the buffers."""
    import waternet_amd.ops as ops
    from waternet_amd.ops.conv import (
        ACT_RELU,
        ConvSpec,
        build_pack_descriptor,
        mark_specs_packed,
    )

    calls = {"n": 0}

    class FakeExt:
        @staticmethod
        def pack_weight_fwd(w, kp, cp):
            calls["n"] += 1
            return torch.zeros(kp * w.shape[2] * w.shape[3] * cp)

        @staticmethod
        def pack_weight_dgrad(w, kp, cp):
            return torch.zeros(cp * w.shape[2] * w.shape[3] * kp)

    monkeypatch.setattr(ops, "_ext", FakeExt())
    monkeypatch.setattr(ops, "_tried", True)
    mods = [torch.nn.Conv2d(3, 5, 3, padding="same"),
            torch.nn.Conv2d(5, 7, 5, padding="same")]
    specs = [ConvSpec(m, ACT_RELU) for m in mods]
    desc = build_pack_descriptor(specs, torch.device("cpu"))
    assert desc.shape == (2, 9) and desc.dtype == torch.int64
    for row, s in zip(desc.tolist(), specs):
        assert row[0] == s.mod.weight.data.data_ptr()
        assert row[1] == s._wp.data_ptr() and row[2] == s._wd.data_ptr()
        assert row[3:] == [s.K, s.C, s.ks, s.ks, s.Kp, s.Cp]

    n_before = calls["n"]
    mark_specs_packed(specs)  # as after ext().pack_all rewrote the buffers
    for s in specs:
        s.refresh_if_needed()  # must SKIP: versions were recorded
    assert calls["n"] == n_before


def test_native_dispatch_policy(monkeypatch):
    """_use_native: never on CPU; on GPU it REFUSES to silently fall back
    to eager when the extension is missing (the loud-failure policy the
    round-end native-code check depends on); WATERNET_AMD_EAGER=1 is the
    only escape."""
    import waternet_amd.ops as ops
    from waternet_amd.models.waternet import _use_native

    class FakeCudaTensor:
        is_cuda = True

    assert _use_native(torch.zeros(1)) is False  # CPU tensor -> eager

    monkeypatch.setattr(ops, "_ext", None)
    monkeypatch.setattr(ops, "_tried", True)
    monkeypatch.setattr(ops, "_load_err", "ImportError('no _C')")
    with pytest.raises(RuntimeError, match="refusing to fall back"):
        _use_native(FakeCudaTensor())
    monkeypatch.setenv("WATERNET_AMD_EAGER", "1")
    assert _use_native(FakeCudaTensor()) is False  # explicit escape only


def test_fused_adam_sync_lr_buffer():
    """sync_lr mirrors param_groups lr into the device-resident buffer the
    graphed k_adam reads — only when it actually changed."""
    from waternet_amd.ops.adam import FusedAdam

    m = torch.nn.Linear(4, 4)
    o = FusedAdam(m.parameters(), lr=1e-3)
    assert abs(float(o.lr_buf.item()) - 1e-3) < 1e-9
    o.sync_lr()  # unchanged: no-op
    assert abs(float(o.lr_buf.item()) - 1e-3) < 1e-9
    o.param_groups[0]["lr"] = 2e-4  # what StepLR does on the host
    o.sync_lr()
    assert abs(float(o.lr_buf.item()) - 2e-4) < 1e-9


def test_native_state_plans():
    """The native execution plans mirror the module trees exactly:
    8 CMG specs (ReLU x7 + Sigmoid), 3x3 refiner specs (ReLU), and the
    VGG plan = 16 convs + 4 pools with ReLUs folded into epilogues."""
    from waternet_amd.engine.native import WaterNetNativeState, VggNativeState
    from waternet_amd.models.vgg import PerceptualModel
    from waternet_amd.models.waternet import WaterNet
    from waternet_amd.ops.conv import ACT_RELU, ACT_SIGMOID

    st = WaterNetNativeState(WaterNet())
    assert len(st.cmg_specs) == 8
    assert [s.act for s in st.cmg_specs] == [ACT_RELU] * 7 + [ACT_SIGMOID]
    assert [s.ks for s in st.cmg_specs] == [7, 5, 3, 1, 7, 5, 3, 3]
    assert [s.K for s in st.cmg_specs] == [128, 128, 128, 64, 64, 64, 64, 3]
    assert set(st.refiner_specs) == {"wb_refiner", "ce_refiner",
                                     "gc_refiner"}
    for specs in st.refiner_specs.values():
        assert [s.ks for s in specs] == [7, 5, 3]
        assert [(s.C, s.K) for s in specs] == [(6, 32), (32, 32), (32, 3)]
        assert all(s.act == ACT_RELU for s in specs)

    vst = VggNativeState(PerceptualModel())
    kinds = [k for k, _ in vst.plan]
    assert kinds.count("conv") == 16 and kinds.count("pool") == 4
    # VGG19-E channel ladder, final pool dropped
    chans = [spec.K for k, spec in vst.plan if k == "conv"]
    assert chans == [64, 64, 128, 128, 256, 256, 256, 256,
                     512, 512, 512, 512, 512, 512, 512, 512]
