"""Production GPU train-step engine — the single step implementation behind
bench.py, train.py's fast path, and __graft_entry__.smoke().

Per step (reference per-minibatch work):
  GPU preprocess (wb/gamma/clahe on the uint8 batch) -> fused uint8 input
  build (cat folding, no NCHW intermediates) -> WaterNet fwd (native MFMA
  kernels, NHWC bf16 end to end) -> VGG-perceptual + MSE loss (fused
  normalize + sqdiff reductions IN NHWC — no layout round trips between
  the towers) -> backward (dgrad/wgrad HIP) -> [DDP: one flat-arena RCCL
  all-reduce overlapped with metrics] -> fused Adam + per-minibatch StepLR
  -> SSIM/PSNR metrics accumulated on-device.

hipGraph capture (use_graph=True): the whole step (preprocess through Adam
and metric accumulation) is captured once and replayed per step, removing
~150 host launch overheads; fresh data is copied into the static input
buffers before each replay so no timed work is skipped. The StepLR lr is
propagated into a device buffer read by the Adam kernel, keeping
per-minibatch LR semantics under replay. Ragged batches (the val split's
last batch, reference train.py:233-235) run the same body eagerly at their
true size via step_batch/eval_batch.
"""

import numpy as np
import torch

from waternet_amd.engine.losses import PERCEPTUAL_WEIGHT
from waternet_amd.engine.native import (
    vgg_forward_nhwc,
    vgg_prepack,
    waternet_forward_from_inputs,
)
from waternet_amd.models.vgg import PerceptualModel
from waternet_amd.models.waternet import WaterNet
from waternet_amd.ops.adam import FusedAdam
from waternet_amd.ops.functional import NormalizeNhwc, mse255_nhwc
from waternet_amd.ops.preprocess import gpu_transform_batch
from waternet_amd.utils.streams import DEBUG as STREAM_DEBUG, StreamJoin
from waternet_amd.ops.ssim import ssim_nhwc

TRAIN_KEYS = ["loss", "perceptual", "mse255", "ssim", "psnr"]
VAL_KEYS = ["mse", "ssim", "psnr", "perceptual_loss"]

# world-invariant VGG seed: the frozen perceptual model must be IDENTICAL
# on every DDP rank regardless of the per-rank data seed (the reference has
# one pretrained VGG; with random init the equivalent is one shared seed).
VGG_SEED = 1234


@torch.no_grad()
def eval_metrics_batch(model, vgg, raw_u8, ref_u8):
    """GPU preprocess + native forward + the VAL metrics for one uint8
    device batch. Returns a float64 4-vector in VAL_KEYS order
    [mse255, ssim, psnr, perceptual] — the building block of both the
    engine's eval epochs and score.py's fast path."""
    from waternet_amd.ops import ext

    wb_u8, gc_u8, he_u8 = gpu_transform_batch(raw_u8)
    e = ext()
    inputs = e.build_inputs_u8(raw_u8, wb_u8, he_u8, gc_u8)
    ref_nhwc = e.u8_to_nhwc(ref_u8, 16)
    fy = vgg_forward_nhwc(vgg, NormalizeNhwc.apply(ref_nhwc))
    out_nhwc = waternet_forward_from_inputs(model, *inputs)
    fx = vgg_forward_nhwc(vgg, NormalizeNhwc.apply(out_nhwc))
    perceptual = mse255_nhwc(fx, fy, 512)
    mse = mse255_nhwc(out_nhwc, ref_nhwc, 3)
    ssim = ssim_nhwc(out_nhwc, ref_nhwc, 3, 1.0)
    mse01 = mse.double() / (255.0 * 255.0)
    psnr = 10.0 * torch.log10(1.0 / mse01)
    return torch.stack([mse.double(), ssim.double(), psnr,
                        perceptual.double()])


class FastStepEngine:
    """GPU train/eval step runner over the native CDNA4 kernel library.

    Owns the FusedAdam flat arena, the frozen VGG tower, the static input
    buffers + hipGraph, and on-device metric accumulators (one host sync
    per epoch, not per batch — the reference's per-batch .item() syncs are
    train.py:136-144)."""

    def __init__(self, model, batch_size=16, height=112, width=112,
                 device="cuda:0", world_size=1, use_graph=True, lr=1e-3,
                 vgg_seed=VGG_SEED):
        self.device = torch.device(device)
        if self.device.type != "cuda":
            raise RuntimeError("FastStepEngine requires a ROCm GPU "
                               "(use the eager trainer on CPU)")
        from waternet_amd.ops import native_available, native_load_error
        if not native_available():
            raise RuntimeError(
                "FastStepEngine requires the native HIP extension: "
                f"{native_load_error()}")
        self.world = world_size
        self.bs = batch_size
        self.h, self.w = height, width
        if height % 8 or width % 8:
            raise RuntimeError("GPU preprocess needs H, W divisible by 8")

        self.model = model
        self.vgg = PerceptualModel(seed=vgg_seed).to(self.device).eval()
        # Pre-pack the 16 frozen VGG conv weights ON THE MAIN STREAM now:
        # the fy tower runs on a side stream and must never be the first
        # (and only) place the packed buffers are written.
        vgg_prepack(self.vgg)
        self.opt = FusedAdam(self.model.parameters(), lr=lr, model=self.model)
        self.sched = torch.optim.lr_scheduler.StepLR(self.opt,
                                                     step_size=10000,
                                                     gamma=0.1)

        # static DEVICE input buffers (graph-capture safe H2D targets)
        self.raw_static = torch.empty(
            (batch_size, height, width, 3), dtype=torch.uint8,
            device=self.device)
        self.ref_static = torch.empty_like(self.raw_static)
        # on-device metric accumulators (TRAIN_KEYS / VAL_KEYS order)
        self.metric_sums = torch.zeros(5, dtype=torch.float64,
                                       device=self.device)
        self.eval_sums = torch.zeros(4, dtype=torch.float64,
                                     device=self.device)
        self._steps = 0
        self._eval_batches = 0
        self._graph = None
        # sanitizer mode (WATERNET_AMD_STREAM_DEBUG=1) synchronizes every
        # cross-stream join, which is illegal inside graph capture — run
        # eager so misordered reads fail loudly instead of racing
        self._use_graph = use_graph and not STREAM_DEBUG
        self._comm_stream = torch.cuda.Stream()
        self._vgg_stream = torch.cuda.Stream()
        self._pack_desc = None
        self._pack_specs = None

    # ---- weight packing -------------------------------------------------
    def _repack_weights(self):
        """Batched fwd+dgrad repack of every WaterNet layer in ONE kernel
        (vs ~36 lazy per-layer pack launches per step). No-op until the
        native ConvSpec table exists (first forward builds it)."""
        if self._pack_desc is None:
            st = getattr(self.model, "_wn_native_state", None)
            if st is None:
                return
            from waternet_amd.ops.conv import build_pack_descriptor

            specs = list(st.cmg_specs)
            for sl in st.refiner_specs.values():
                specs.extend(sl)
            self._pack_specs = specs
            self._pack_desc = build_pack_descriptor(specs, self.device)
        from waternet_amd.ops import ext
        from waternet_amd.ops.conv import mark_specs_packed

        ext().pack_all(self._pack_desc, len(self._pack_specs))
        mark_specs_packed(self._pack_specs)

    # ---- one full training step on the current stream -------------------
    def _build_all_inputs(self, raw_u8, ref_u8):
        """uint8 batch -> (conv inputs 4-tuple, ref NHWC bf16)."""
        from waternet_amd.ops import ext

        wb_u8, gc_u8, he_u8 = gpu_transform_batch(raw_u8)
        e = ext()
        # he fills the `ce` slot (reference train.py:108 -> net.py:99)
        inputs = e.build_inputs_u8(raw_u8, wb_u8, he_u8, gc_u8)
        ref_nhwc = e.u8_to_nhwc(ref_u8, 16)
        return inputs, ref_nhwc

    def _run_step_body(self, raw_u8, ref_u8):
        from waternet_amd.utils.profiling import trace_range

        self._repack_weights()
        with trace_range("preprocess"):
            inputs, ref_nhwc = self._build_all_inputs(raw_u8, ref_u8)

        # The fy tower depends only on ref: run it on a second stream
        # concurrently with the WaterNet forward + fx tower (the deep VGG
        # layers underfill the 256 CUs, so the towers overlap). StreamJoin
        # asserts the producer ordering (sanitizer mode forces completion).
        self._vgg_stream.wait_stream(torch.cuda.current_stream())
        vgg_join = StreamJoin(self._vgg_stream)
        with torch.cuda.stream(self._vgg_stream), torch.no_grad(), \
                trace_range("vgg_ref"):
            fy = vgg_forward_nhwc(self.vgg, NormalizeNhwc.apply(ref_nhwc))
        vgg_join.mark()

        with trace_range("forward"):
            out_nhwc = waternet_forward_from_inputs(self.model, *inputs)

        with trace_range("loss"):
            fx = vgg_forward_nhwc(self.vgg, NormalizeNhwc.apply(out_nhwc))
            vgg_join.wait()
            perceptual = mse255_nhwc(fx, fy, 512)
            mse = mse255_nhwc(out_nhwc, ref_nhwc, 3)
            loss = PERCEPTUAL_WEIGHT * perceptual + mse

        with trace_range("backward"):
            self.opt.zero_grad()
            loss.backward()
        if self.world > 1:
            # Launch the flat-arena all-reduce on the RCCL/comm stream and
            # overlap it with the SSIM/PSNR metric math on the compute
            # stream (metrics depend only on `out`, not on gradients) —
            # latency-bound 4 MB collective hidden under
            # independent compute.
            with trace_range("allreduce"):
                self.opt.grads.div_(self.world)
                self._comm_stream.wait_stream(torch.cuda.current_stream())
                comm_join = StreamJoin(self._comm_stream)
                with torch.cuda.stream(self._comm_stream):
                    torch.distributed.all_reduce(self.opt.grads)
                comm_join.mark()

        with trace_range("metrics"), torch.no_grad():
            out_d = out_nhwc.detach()
            ssim = ssim_nhwc(out_d, ref_nhwc, 3, 1.0)
            mse01 = mse.detach().double() / (255.0 * 255.0)
            psnr = 10.0 * torch.log10(1.0 / mse01)
            self.metric_sums += torch.stack([
                loss.detach().double(), perceptual.detach().double(),
                mse.detach().double(), ssim.double(), psnr
            ])

        if self.world > 1:
            comm_join.wait()
        with trace_range("optimizer"):
            self.opt.step()

    # ---- graph capture ---------------------------------------------------
    def _maybe_capture(self):
        if self._graph is not None or not self._use_graph:
            return
        # The 2 warmup executions advance the optimizer/metric state; keep
        # capture side-effect-free by snapshotting and restoring it so the
        # first replay computes exactly the step an eager run would.
        o = self.opt
        snap = {
            "master": o.master.clone(), "grads": o.grads.clone(),
            "exp_avg": o.exp_avg.clone(), "exp_avg_sq": o.exp_avg_sq.clone(),
            "step_buf": o.step_buf.clone(), "metric": self.metric_sums.clone(),
        }
        host_step = o._step
        # warmup on a side stream, then capture one step
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._run_step_body(self.raw_static, self.ref_static)
        torch.cuda.current_stream().wait_stream(s)
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._run_step_body(self.raw_static, self.ref_static)
            self._graph = g
        except Exception as e:  # noqa: BLE001
            import sys

            print(f"[fast] hipGraph capture failed, running eager: {e!r}",
                  file=sys.stderr)
            self._use_graph = False
        # restore pre-warmup state (capture itself records, not executes)
        o.master.copy_(snap["master"])
        o.grads.copy_(snap["grads"])
        o.exp_avg.copy_(snap["exp_avg"])
        o.exp_avg_sq.copy_(snap["exp_avg_sq"])
        o.step_buf.copy_(snap["step_buf"])
        self.metric_sums.copy_(snap["metric"])
        o._step = host_step
        o._mark_model_dirty()  # packed-weight caches reflect restored masters

    # ---- public step API -------------------------------------------------
    def load_batch(self, raw_host_u8, ref_host_u8):
        """Async H2D copy of a (bs,H,W,3) uint8 pair into the static
        buffers (pin the host tensors for a true async copy)."""
        self.raw_static.copy_(raw_host_u8, non_blocking=True)
        self.ref_static.copy_(ref_host_u8, non_blocking=True)

    def step(self):
        """One training step on the data currently in the static buffers
        (graph replay at steady state)."""
        if self._use_graph:
            self._maybe_capture()
        # per-minibatch StepLR semantics: host scheduler updates lr; the
        # fused Adam reads it from param_groups (eager) / device buffer
        # (graph path propagates below).
        if self._graph is not None:
            self.opt.sync_lr()  # propagate StepLR changes to the device buf
            self._graph.replay()
            self.opt._step += 1  # host mirror of the captured device tick
        else:
            self._run_step_body(self.raw_static, self.ref_static)
        self.sched.step()
        self._steps += 1

    def step_batch(self, raw_u8, ref_u8):
        """Eager step on arbitrary-size device uint8 tensors (ragged last
        batch of an epoch)."""
        self._run_step_body(raw_u8, ref_u8)
        self.sched.step()
        self._steps += 1

    # ---- evaluation -------------------------------------------------------
    @torch.no_grad()
    def eval_batch(self, raw_u8, ref_u8):
        """No-grad forward + metrics on a device uint8 batch; accumulates
        into eval_sums (VAL_KEYS order). Any batch size."""
        self.eval_sums += eval_metrics_batch(self.model, self.vgg, raw_u8,
                                             ref_u8)
        self._eval_batches += 1

    def reset_eval(self):
        self.eval_sums.zero_()
        self._eval_batches = 0

    def eval_metrics(self):
        """One host sync: averaged VAL metrics since reset_eval()."""
        n = max(self._eval_batches, 1)
        vals = (self.eval_sums / n).tolist()
        return dict(zip(VAL_KEYS, vals))

    # ---- metric accounting -----------------------------------------------
    def metrics_snapshot(self):
        """Device-side snapshot for per-epoch accounting (no host sync)."""
        return self.metric_sums.clone(), self._steps

    def metrics_since(self, snap):
        """One host sync: averaged TRAIN metrics since `snap`."""
        sums0, n0 = snap
        n = max(self._steps - n0, 1)
        vals = ((self.metric_sums - sums0) / n).tolist()
        return dict(zip(TRAIN_KEYS, vals))

    def metrics(self):
        n = max(self._steps, 1)
        vals = (self.metric_sums / n).tolist()
        return dict(zip(TRAIN_KEYS, vals))


class BenchTrainer(FastStepEngine):
    """FastStepEngine + a deterministic synthetic pinned-host data pool
    (the bench/smoke entry; BASELINE config 2). Every step pays the real
    host->HBM transfer the reference's dataloader pays."""

    def __init__(self, batch_size=16, height=112, width=112, device="cuda:0",
                 world_size=1, seed=0, use_graph=True, pool_size=4,
                 lr=1e-3):
        torch.manual_seed(seed)
        model = WaterNet().to(torch.device(device))
        super().__init__(model, batch_size=batch_size, height=height,
                         width=width, device=device, world_size=world_size,
                         use_graph=use_graph, lr=lr)
        # synthetic uint8 data pool (deterministic per rank), kept in PINNED
        # HOST memory. The images are spatially-correlated noise (low-res
        # noise bilinearly upsampled + fine noise), matching natural-image
        # statistics rather than full-range white noise — the reference
        # trains on photographs.
        rng = np.random.default_rng(seed)

        def synth_images():
            base = rng.integers(
                0, 256,
                size=(batch_size, max(height // 8, 1), max(width // 8, 1), 3),
            ).astype(np.float32)
            t = torch.from_numpy(base).permute(0, 3, 1, 2)
            up = torch.nn.functional.interpolate(
                t, size=(height, width), mode="bilinear", align_corners=False
            ).permute(0, 2, 3, 1).numpy()
            fine = rng.normal(0.0, 12.0, size=up.shape)
            return torch.from_numpy(
                np.clip(up + fine, 0, 255).astype(np.uint8))

        self.pool = []
        for _ in range(pool_size):
            raw, ref = synth_images(), synth_images()
            self.pool.append((raw.pin_memory(), ref.pin_memory()))
        self._i = 0

    def step(self):
        raw, ref = self.pool[self._i % len(self.pool)]
        self._i += 1
        self.load_batch(raw, ref)
        super().step()
