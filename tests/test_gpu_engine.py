"""GPU integration tests: bench trainer, hipGraph capture, inference engine,
postprocess bridges."""

import numpy as np
import pytest
import torch

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


def test_u8_bridges():
    from waternet_amd.ops import ext

    e = ext()
    rng = np.random.default_rng(0)
    raw = torch.from_numpy(
        rng.integers(0, 256, size=(2, 16, 16, 3), dtype=np.uint8)
    ).to(DEV)
    f = e.u8_to_nchw(raw)
    assert f.shape == (2, 3, 16, 16)
    ref = raw.permute(0, 3, 1, 2).float() / 255.0
    assert torch.allclose(f, ref, atol=1e-6)

    # out_to_u8: clip*255 trunc on NHWC bf16
    x = torch.rand(2, 16, 16, 16, device=DEV).bfloat16()
    u = e.out_to_u8(x)
    ref_u = (x[..., :3].float().clamp(0, 1) * 255).to(torch.uint8)
    assert (u.int() - ref_u.int()).abs().max().item() <= 1


def test_bench_trainer_eager_steps():
    from waternet_amd.engine.fast import BenchTrainer

    tr = BenchTrainer(batch_size=2, height=64, width=64, device=DEV,
                      use_graph=False, seed=3)
    for _ in range(3):
        tr.step()
    torch.cuda.synchronize()
    m = tr.metrics()
    assert all(np.isfinite(v) for v in m.values()), m
    assert m["ssim"] <= 1.0


def test_bench_trainer_graph_matches_eager():
    """hipGraph-captured step must track the eager step (same seeds/data)."""
    from waternet_amd.engine.fast import BenchTrainer

    tr_e = BenchTrainer(batch_size=2, height=64, width=64, device=DEV,
                        use_graph=False, seed=7)
    tr_g = BenchTrainer(batch_size=2, height=64, width=64, device=DEV,
                        use_graph=True, seed=7)
    for _ in range(4):
        tr_e.step()
        tr_g.step()
    torch.cuda.synchronize()
    me, mg = tr_e.metrics(), tr_g.metrics()
    if tr_g._graph is None:
        pytest.skip("graph capture unavailable on this stack")
    for k in me:
        assert abs(me[k] - mg[k]) / (abs(me[k]) + 1e-6) < 5e-2, (k, me, mg)


def test_inference_engine_matches_cpu_reference():
    """GPU frame pipeline vs the CPU numpy transforms + eager model."""
    import os

    from waternet_amd.data.bridge import arr2ten, ten2arr
    from waternet_amd.data.transforms import transform
    from waternet_amd.engine.inferencer import InferenceEngine
    from waternet_amd.models.waternet import WaterNet

    torch.manual_seed(11)
    model = WaterNet().to(DEV)
    rng = np.random.default_rng(11)
    frame = rng.integers(0, 256, size=(64, 64, 3), dtype=np.uint8)

    eng = InferenceEngine(model, 64, 64, device=DEV, use_graph=True)
    got = eng.infer_frame(frame)
    got2 = eng.infer_frame(frame)  # replay path
    assert np.array_equal(got, got2)

    # CPU reference (eager fp32 model on CPU transforms)
    os.environ["WATERNET_AMD_EAGER"] = "1"
    try:
        wb, gc, he = transform(frame)
        with torch.no_grad():
            out = model(
                arr2ten(frame, True).to(DEV), arr2ten(wb, True).to(DEV),
                arr2ten(he, True).to(DEV), arr2ten(gc, True).to(DEV)
            )
        ref = ten2arr(out)[0]
    finally:
        os.environ.pop("WATERNET_AMD_EAGER")
    diff = np.abs(got.astype(int) - ref.astype(int))
    # bf16 conv path + uint8 rounding: allow small differences
    assert np.mean(diff) < 2.0 and np.max(diff) <= 16, (
        diff.mean(), diff.max())


def test_highres_train_step():
    """BASELINE config 5 path: one 512x512 training step through the native
    kernels (small batch here; bs=64/GPU is the 8-GPU stress config)."""
    from waternet_amd.engine.fast import BenchTrainer

    tr = BenchTrainer(batch_size=2, height=512, width=512, device=DEV,
                      use_graph=False, seed=13, pool_size=1)
    tr.step()
    torch.cuda.synchronize()
    m = tr.metrics()
    assert all(np.isfinite(v) for v in m.values()), m


def test_inference_engine_1080p_shape():
    """BASELINE config 4 path: one 1080p frame through the hipGraph engine."""
    from waternet_amd.engine.inferencer import InferenceEngine
    from waternet_amd.models.waternet import WaterNet

    torch.manual_seed(17)
    model = WaterNet().to(DEV)
    eng = InferenceEngine(model, 1088, 1920, device=DEV, use_graph=True)
    rng = np.random.default_rng(17)
    frame = rng.integers(0, 256, size=(1088, 1920, 3), dtype=np.uint8)
    out = eng.infer_frame(frame)
    assert out.shape == (1088, 1920, 3) and out.dtype == np.uint8


def test_preprocess_112():
    """Full-size preprocess at the flagship 112x112 shape."""
    from waternet_amd.ops.preprocess import gpu_transform_batch

    rng = np.random.default_rng(5)
    raw = torch.from_numpy(
        rng.integers(0, 256, size=(16, 112, 112, 3), dtype=np.uint8)
    ).to(DEV)
    wb, gc, he = gpu_transform_batch(raw)
    torch.cuda.synchronize()
    for t in (wb, gc, he):
        assert t.shape == raw.shape and t.dtype == torch.uint8


def test_stream_join_ordering():
    """Race-assertion helpers: copy-stream produce -> compute consume."""
    from waternet_amd.utils.streams import StreamJoin, ordered_copy

    cs = torch.cuda.Stream()
    src = torch.randn(1 << 20, device=DEV)
    dst = torch.empty_like(src)
    ev = ordered_copy(dst, src, cs)
    torch.cuda.current_stream().wait_event(ev)
    assert torch.equal(dst, src)

    join = StreamJoin(cs)
    with torch.cuda.stream(cs):
        dst.add_(1.0)
    join.mark()
    join.wait()
    torch.cuda.synchronize()
    assert torch.allclose(dst, src + 1.0)

    fresh = StreamJoin(cs)
    with pytest.raises(AssertionError):
        fresh.wait()  # wait() without mark() must fail loudly


# ---------------------------------------------------------------------------
# Round 2: unified fast engine (the train.py production path)
# ---------------------------------------------------------------------------


def test_fast_engine_loss_matches_eager_fp32():
    """One eager-mode FastStepEngine step's loss vs the same computation in
    plain PyTorch fp32 (same weights, same preprocessed inputs): validates
    the full-NHWC wiring (build_inputs_u8 -> convs -> NormalizeNhwc -> VGG
    -> Mse255) against the reference composition."""
    import os

    from waternet_amd.engine.fast import FastStepEngine
    from waternet_amd.engine.losses import PERCEPTUAL_WEIGHT
    from waternet_amd.engine.native import (
        vgg_forward_nhwc,
        waternet_forward_from_inputs,
    )
    from waternet_amd.models.vgg import normalize_imagenet
    from waternet_amd.models.waternet import WaterNet
    from waternet_amd.ops.functional import NormalizeNhwc, mse255_nhwc

    torch.manual_seed(21)
    model = WaterNet().to(DEV)
    eng = FastStepEngine(model, batch_size=2, height=64, width=64,
                         device=DEV, use_graph=False)
    rng = np.random.default_rng(3)
    raw = torch.from_numpy(
        rng.integers(0, 256, size=(2, 64, 64, 3), dtype=np.uint8)).to(DEV)
    ref = torch.from_numpy(
        rng.integers(0, 256, size=(2, 64, 64, 3), dtype=np.uint8)).to(DEV)

    inputs, ref_nhwc = eng._build_all_inputs(raw, ref)
    with torch.no_grad():
        out_nhwc = waternet_forward_from_inputs(model, *inputs)
        fx = vgg_forward_nhwc(eng.vgg, NormalizeNhwc.apply(out_nhwc))
        fy = vgg_forward_nhwc(eng.vgg, NormalizeNhwc.apply(ref_nhwc))
        perceptual = mse255_nhwc(fx, fy, 512)
        mse = mse255_nhwc(out_nhwc, ref_nhwc, 3)
        loss_native = (PERCEPTUAL_WEIGHT * perceptual + mse).item()

    # eager fp32 reference on the SAME preprocessed inputs
    cmg_in = inputs[0]
    raw_f = cmg_in[..., 0:3].permute(0, 3, 1, 2).float()
    wb_f = cmg_in[..., 3:6].permute(0, 3, 1, 2).float()
    he_f = cmg_in[..., 6:9].permute(0, 3, 1, 2).float()
    gc_f = cmg_in[..., 9:12].permute(0, 3, 1, 2).float()
    ref_f = ref_nhwc[..., :3].permute(0, 3, 1, 2).float()
    os.environ["WATERNET_AMD_EAGER"] = "1"
    try:
        with torch.no_grad():
            out_e = model(raw_f, wb_f, he_f, gc_f)
            fx_e = eng.vgg(normalize_imagenet(out_e))
            fy_e = eng.vgg(normalize_imagenet(ref_f))
            dp = 255.0 * (fx_e - fy_e)
            dm = 255.0 * (out_e - ref_f)
            loss_eager = (PERCEPTUAL_WEIGHT * torch.mean(dp * dp)
                          + torch.mean(dm * dm)).item()
    finally:
        os.environ.pop("WATERNET_AMD_EAGER", None)
    rel = abs(loss_native - loss_eager) / max(abs(loss_eager), 1e-9)
    assert rel < 0.05, (loss_native, loss_eager, rel)


def test_fast_engine_ragged_batches():
    """step_batch / eval_batch at a size different from the static batch
    (the reference val loader's last batch is ragged, train.py:233-235)."""
    from waternet_amd.engine.fast import FastStepEngine
    from waternet_amd.models.waternet import WaterNet

    torch.manual_seed(2)
    model = WaterNet().to(DEV)
    eng = FastStepEngine(model, batch_size=4, height=64, width=64,
                         device=DEV, use_graph=False)
    rng = np.random.default_rng(1)

    def batch(n):
        return (torch.from_numpy(rng.integers(
                    0, 256, size=(n, 64, 64, 3), dtype=np.uint8)).to(DEV),
                torch.from_numpy(rng.integers(
                    0, 256, size=(n, 64, 64, 3), dtype=np.uint8)).to(DEV))

    raw, ref = batch(4)
    eng.load_batch(raw, ref)
    eng.step()
    eng.step_batch(*batch(3))  # ragged train tail
    eng.eval_batch(*batch(4))
    eng.eval_batch(*batch(2))  # ragged val tail
    torch.cuda.synchronize()
    m = eng.metrics()
    assert all(np.isfinite(v) for v in m.values()), m
    ev = eng.eval_metrics()
    assert all(np.isfinite(v) for v in ev.values()), ev
    assert eng._eval_batches == 2


def test_train_cli_fast_gpu(tmp_path):
    """train.py end-to-end on GPU through the fast engine: synthetic data,
    2 epochs, graph capture + ragged batches, reference-format outputs."""
    import json
    import os

    import train as train_cli

    os.environ["WATERNET_TRAINING_DIR"] = str(tmp_path)
    try:
        train_cli.main([
            "--epochs", "2", "--batch-size", "8", "--height", "64",
            "--width", "64", "--synthetic", "24", "--full-state",
        ])
    finally:
        os.environ.pop("WATERNET_TRAINING_DIR", None)
    savedir = tmp_path / "0"
    sd = torch.load(savedir / "last.pt", map_location="cpu")
    from waternet_amd.models.waternet import WaterNet

    m = WaterNet()
    m.load_state_dict(sd)  # byte-compatible 34-tensor schema
    assert len(sd) == 34
    train_csv = (savedir / "metrics-train.csv").read_text().splitlines()
    assert train_csv[0] == "mse,ssim,psnr,perceptual_loss,loss"
    assert len(train_csv) == 3  # header + 2 epochs
    cfg = json.loads((savedir / "config.json").read_text())
    assert cfg["batch_size"] == 8
    assert (savedir / "last-trainstate.pt").exists()


def test_train_cli_fast_gpu_disk_dataset(tmp_path):
    """The fast engine path with a REAL on-disk image dataset (PIL decode
    in dataloader workers -> uint8 raw_mode batches -> GPU preprocess
    in-step), matching the reference's raw-890/reference-890 layout."""
    from PIL import Image

    rng = np.random.default_rng(7)
    (tmp_path / "data" / "raw-890").mkdir(parents=True)
    (tmp_path / "data" / "reference-890").mkdir(parents=True)
    for i in range(8):
        for d in ("raw-890", "reference-890"):
            Image.fromarray(rng.integers(
                0, 256, size=(80, 80, 3), dtype=np.uint8
            )).save(tmp_path / "data" / d / f"{i}.png")
    from waternet_amd.data.dataset import UIEBDataset
    from waternet_amd.engine.fast import FastStepEngine
    from waternet_amd.models.waternet import WaterNet

    ds = UIEBDataset(tmp_path / "data" / "raw-890",
                     tmp_path / "data" / "reference-890",
                     im_height=64, im_width=64, raw_mode=True)
    loader = torch.utils.data.DataLoader(ds, batch_size=4, num_workers=2,
                                         pin_memory=True)
    torch.manual_seed(0)
    model = WaterNet().to(DEV)
    eng = FastStepEngine(model, batch_size=4, height=64, width=64,
                         device=DEV, use_graph=True)
    for batch in loader:
        eng.load_batch(batch["raw"], batch["ref"])
        eng.step()
    torch.cuda.synchronize()
    m = eng.metrics()
    assert all(np.isfinite(v) for v in m.values()), m
    assert eng._steps == 2


def test_serve_enhance_gpu():
    """serve.Server.enhance on GPU goes through the hipGraph inference
    engine (per-resolution cache) and returns a valid frame."""
    from serve import Server

    srv = Server(device=DEV)
    rng = np.random.default_rng(3)
    img = rng.integers(0, 256, size=(64, 96, 3), dtype=np.uint8)
    out = srv.enhance(img)
    assert out.shape == (64, 96, 3) and out.dtype == np.uint8
    assert (64, 96) in srv._engines
    out2 = srv.enhance(img)  # cached-engine path
    assert np.array_equal(out, out2)
    # odd sizes run the GPU engine via reflect-pad + crop (cache keyed
    # on the padded /8 size)
    img2 = rng.integers(0, 256, size=(33, 45, 3), dtype=np.uint8)
    out3 = srv.enhance(img2)
    assert out3.shape == (33, 45, 3)
    assert (40, 48) in srv._engines
    # dimension cap enforced
    import serve as serve_mod

    big = np.zeros((serve_mod.MAX_DIM + 8, 8, 3), dtype=np.uint8)
    import pytest as _pytest

    with _pytest.raises(ValueError):
        srv.enhance(big)


def test_engine_step_sanitizer_mode(monkeypatch):
    """One real engine step with WATERNET_AMD_STREAM_DEBUG semantics: every
    cross-stream join synchronizes (misordered reads become deterministic
    failures) and graph capture is disabled."""
    import waternet_amd.engine.fast as fastmod
    import waternet_amd.utils.streams as streams

    monkeypatch.setattr(streams, "DEBUG", True)
    monkeypatch.setattr(fastmod, "STREAM_DEBUG", True)
    from waternet_amd.models.waternet import WaterNet

    torch.manual_seed(4)
    model = WaterNet().to(DEV)
    eng = fastmod.FastStepEngine(model, batch_size=2, height=64, width=64,
                                 device=DEV, use_graph=True)
    assert eng._use_graph is False  # sanitizer forces eager
    rng = np.random.default_rng(0)
    raw = torch.from_numpy(rng.integers(0, 256, (2, 64, 64, 3),
                                        dtype=np.uint8))
    eng.load_batch(raw.to(DEV), raw.to(DEV))
    eng.step()
    eng.step()
    torch.cuda.synchronize()
    assert all(np.isfinite(v) for v in eng.metrics().values())


def test_score_cli_fast_vs_eager(tmp_path):
    """score.py's fast path (GPU preprocess + native eval) agrees with the
    eager reference composition on the same checkpoint/split."""
    import score as score_cli

    torch.manual_seed(0)
    from waternet_amd.models.waternet import WaterNet

    ckpt = tmp_path / "w.pt"
    torch.save(WaterNet().state_dict(), ckpt)
    common = ["--weights", str(ckpt), "--synthetic", "40",
              "--batch-size", "8", "--height", "64", "--width", "64"]
    m_fast = score_cli.main(common + ["--engine", "fast"])
    m_eager = score_cli.main(common + ["--engine", "eager"])
    for k in ("mse", "ssim", "psnr"):
        a, b = m_fast[k], m_eager[k]
        assert abs(a - b) / (abs(b) + 1e-6) < 0.08, (k, a, b)
    # perceptual uses DIFFERENT random VGG draws only if seeding diverged;
    # both paths build PerceptualModel() after identical RNG history, so
    # it must agree too
    a, b = m_fast["perceptual_loss"], m_eager["perceptual_loss"]
    assert abs(a - b) / (abs(b) + 1e-6) < 0.10, (a, b)
