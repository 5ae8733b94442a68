"""Benchmark/production train-step runner (single stream of truth for
bench.py and the hipGraph-captured fast path).

Per step (reference per-minibatch work):
  GPU preprocess (wb/gamma/clahe) -> input build -> WaterNet fwd (native
  MFMA kernels) -> VGG-perceptual + MSE loss -> backward (dgrad/wgrad HIP)
  -> [DDP: one flat-arena RCCL all-reduce] -> fused Adam + per-minibatch
  StepLR -> SSIM/PSNR metrics accumulated on-device.

hipGraph capture (use_graph=True): the whole step (preprocess through Adam
and metric accumulation) is captured once and replayed per step, removing
~150 host launch overheads; fresh synthetic data is copied into the static
input buffers before each replay so no timed work is skipped. The StepLR
lr is propagated into a device buffer read by the Adam kernel, keeping
per-minibatch LR semantics under replay.
"""

import numpy as np
import torch

from waternet_amd.engine.losses import PERCEPTUAL_WEIGHT
from waternet_amd.models.vgg import PerceptualModel, normalize_imagenet
from waternet_amd.models.waternet import WaterNet
from waternet_amd.ops.adam import FusedAdam
from waternet_amd.ops.preprocess import gpu_transform_batch
from waternet_amd.ops.ssim import ssim_native


class BenchTrainer:
    def __init__(self, batch_size=16, height=112, width=112, device="cuda:0",
                 world_size=1, seed=0, use_graph=True, pool_size=4,
                 lr=1e-3):
        self.device = torch.device(device)
        self.world = world_size
        self.bs = batch_size

        torch.manual_seed(seed)
        self.model = WaterNet().to(self.device)
        # common seed: the frozen VGG must be identical on every DDP rank
        self.vgg = PerceptualModel(seed=1234).to(self.device).eval()
        self.opt = FusedAdam(self.model.parameters(), lr=lr, model=self.model)
        self.sched = torch.optim.lr_scheduler.StepLR(self.opt,
                                                     step_size=10000,
                                                     gamma=0.1)
        # synthetic uint8 data pool (deterministic per rank), kept in PINNED
        # HOST memory so every step pays the real host->HBM transfer the
        # reference's dataloader pays. The images are
        # spatially-correlated noise (low-res noise bilinearly upsampled +
        # fine noise), matching natural-image statistics rather than
        # full-range white noise — the reference trains on photographs.
        rng = np.random.default_rng(seed)
        pin = self.device.type == "cuda"

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
            if pin:
                raw, ref = raw.pin_memory(), ref.pin_memory()
            self.pool.append((raw, ref))
        # static DEVICE input buffers (graph-capture safe H2D targets)
        self.raw_static = torch.empty(
            (batch_size, height, width, 3), dtype=torch.uint8,
            device=self.device)
        self.ref_static = torch.empty_like(self.raw_static)
        # on-device metric accumulators
        self.metric_sums = torch.zeros(5, dtype=torch.float64,
                                       device=self.device)
        self._i = 0
        self._graph = None
        self._use_graph = use_graph
        self._comm_stream = (torch.cuda.Stream()
                             if self.device.type == "cuda" else None)
        self._vgg_stream = (torch.cuda.Stream()
                            if self.device.type == "cuda" else None)
        self._pack_desc = None
        self._pack_specs = None

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

    # ---- one full training step on the current stream ----
    def _run_step_body(self):
        from waternet_amd.utils.profiling import trace_range

        if self.device.type == "cuda":
            self._repack_weights()
        raw_u8, ref_u8 = self.raw_static, self.ref_static
        with trace_range("preprocess"):
            wb_u8, gc_u8, he_u8 = gpu_transform_batch(raw_u8)

            from waternet_amd.ops import ext

            e = ext()
            raw_f = e.u8_to_nchw(raw_u8)
            wb_f = e.u8_to_nchw(wb_u8)
            gc_f = e.u8_to_nchw(gc_u8)
            he_f = e.u8_to_nchw(he_u8)
            ref_f = e.u8_to_nchw(ref_u8)

        # The fy tower depends only on ref: run it on a second stream
        # concurrently with the WaterNet forward + fx tower (the deep VGG
        # layers underfill the 256 CUs, so the towers overlap).
        if self._vgg_stream is not None:
            self._vgg_stream.wait_stream(torch.cuda.current_stream())
            with torch.cuda.stream(self._vgg_stream), torch.no_grad(), \
                    trace_range("vgg_ref"):
                fy = self.vgg(normalize_imagenet(ref_f))

        with trace_range("forward"):
            out = self.model(raw_f, wb_f, he_f, gc_f)  # he in the ce slot

        with trace_range("loss"):
            fx = self.vgg(normalize_imagenet(out))
            if self._vgg_stream is not None:
                torch.cuda.current_stream().wait_stream(self._vgg_stream)
            else:
                with torch.no_grad():
                    fy = self.vgg(normalize_imagenet(ref_f))
            dp = 255.0 * (fx - fy)
            perceptual = torch.mean(dp * dp)
            dm = 255.0 * (out - ref_f)
            mse = torch.mean(dm * dm)
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
                if self._comm_stream is not None:
                    self._comm_stream.wait_stream(torch.cuda.current_stream())
                    with torch.cuda.stream(self._comm_stream):
                        torch.distributed.all_reduce(self.opt.grads)
                else:  # CPU/gloo path (tests)
                    torch.distributed.all_reduce(self.opt.grads)

        with trace_range("metrics"), torch.no_grad():
            ssim = ssim_native(out.detach(), ref_f, 1.0)
            mse01 = torch.mean((out.detach() - ref_f) ** 2)
            psnr = 10.0 * torch.log10(1.0 / mse01)
            self.metric_sums += torch.stack([
                loss.detach().double(), perceptual.detach().double(),
                mse.detach().double(), ssim.double(), psnr.double()
            ])

        if self.world > 1 and self._comm_stream is not None:
            torch.cuda.current_stream().wait_stream(self._comm_stream)
        with trace_range("optimizer"):
            self.opt.step()

    def _load_batch(self):
        raw, ref = self.pool[self._i % len(self.pool)]
        self.raw_static.copy_(raw, non_blocking=True)  # pinned H2D
        self.ref_static.copy_(ref, non_blocking=True)
        self._i += 1

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
                self._run_step_body()
        torch.cuda.current_stream().wait_stream(s)
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._run_step_body()
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

    def step(self):
        self._load_batch()
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
            self._run_step_body()
        self.sched.step()

    def metrics(self):
        n = max(self._i, 1)
        vals = (self.metric_sums / n).tolist()
        return dict(zip(["loss", "perceptual", "mse255", "ssim", "psnr"],
                        vals))
