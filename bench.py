"""Benchmark entry point (driver contract).

Measures the flagship workload: WaterNet training, bs=16/GPU, 112x112, bf16
compute, on synthetic data with random-init weights — the BASELINE.json
metric ("train images/sec whole node at 112x112 bs=16/GPU"). Each timed step
performs the full reference per-minibatch work: preprocess
(white balance + gamma + CLAHE, GPU-native), forward, VGG-perceptual + MSE
loss, backward, Adam + StepLR, and the SSIM/PSNR minibatch metrics
(accumulated on-device; the reference's per-step .item() syncs are not
replicated inside the timed loop).

Usage: python bench.py --gpus N --steps K --warmup W
For N>1 the driver launches this under torch.distributed.run with one rank
per GPU (RCCL); `value` is the whole-job aggregate images/sec.
"""

import argparse
import json
import os
import time

import torch

BASELINE_IMG_PER_SEC = 12.8  # reference epoch-400 throughput


def parse_args():
    p = argparse.ArgumentParser()
    p.add_argument("--gpus", type=int, default=1)
    p.add_argument("--steps", type=int, default=30)
    p.add_argument("--warmup", type=int, default=10)
    p.add_argument("--batch-size", type=int, default=16)
    p.add_argument("--height", type=int, default=112)
    p.add_argument("--width", type=int, default=112)
    p.add_argument("--no-graph", action="store_true",
                   help="disable hipGraph capture of the train step")
    p.add_argument("--mode", choices=["train", "infer1080p"], default="train",
                   help="train = flagship training step (driver metric); "
                        "infer1080p = BASELINE config 4, hipGraph-captured "
                        "bs=1 1080p per-frame video inference")
    return p.parse_args()


def infer1080p(args):
    """BASELINE config 4: 1080p (1920x1088, /32-padded as the reference
    requires multiples of 32 — training_utils.py:99-103) bs=1 per-frame
    pipeline, hipGraph-captured: H2D -> GPU preprocess -> forward -> fused
    u8 postprocess -> D2H."""
    import numpy as np

    from waternet_amd.engine.inferencer import InferenceEngine
    from waternet_amd.models.waternet import WaterNet

    h, w = 1088, 1920
    torch.manual_seed(0)
    model = WaterNet().to("cuda:0")
    eng = InferenceEngine(model, h, w, device="cuda:0",
                          use_graph=not args.no_graph)
    rng = np.random.default_rng(0)
    frames = [rng.integers(0, 256, size=(h, w, 3), dtype=np.uint8)
              for _ in range(4)]
    for i in range(max(args.warmup, 2)):
        eng.infer_frame(frames[i % len(frames)])
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for i in range(args.steps):
        eng.infer_frame(frames[i % len(frames)])
    torch.cuda.synchronize()
    dt = time.perf_counter() - t0
    print(json.dumps({
        "metric": "video inference frames/sec at 1080p bs=1 (hipGraph)",
        "value": args.steps / dt,
        "unit": "frames/sec",
        "n_gpus": 1,
        "steps": args.steps,
        "warmup": args.warmup,
        "ms_per_step": dt / args.steps * 1000.0,
        "higher_is_better": True,
        "scaling": "weak",
        "vs_baseline": None,
        "dtype": "bf16",
        "data": "synthetic",
        "config": {"model": "waternet", "global_batch": 1,
                   "im_size": f"{h}x{w}", "parallelism": "single",
                   "graph": not args.no_graph},
    }))


def main():
    args = parse_args()
    rank = int(os.environ.get("RANK", 0))
    world = int(os.environ.get("WORLD_SIZE", 1))
    local_rank = int(os.environ.get("LOCAL_RANK", rank))

    if not torch.cuda.is_available():
        raise SystemExit("bench.py requires a ROCm GPU")

    if args.mode == "infer1080p":
        infer1080p(args)
        return

    torch.cuda.set_device(local_rank)
    device = torch.device("cuda", local_rank)

    if world > 1:
        os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
        os.environ.setdefault("MASTER_PORT", "29500")
        torch.distributed.init_process_group("nccl", rank=rank,
                                             world_size=world)

    from waternet_amd.engine.fast import BenchTrainer

    torch.manual_seed(1234 + rank)
    trainer = BenchTrainer(
        batch_size=args.batch_size,
        height=args.height,
        width=args.width,
        device=device,
        world_size=world,
        seed=1234 + rank,
        # The step is GPU-bound (graph vs eager measured equal at bs=16
        # 112^2); keep the RCCL collective out of graph capture under DDP.
        use_graph=(not args.no_graph) and world == 1,
    )

    if world > 1:
        # all ranks start from rank0's params (one-time, untimed)
        torch.distributed.broadcast(trainer.opt.master, src=0)

    for _ in range(max(args.warmup, 1)):
        trainer.step()

    if world > 1:
        torch.distributed.barrier()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(args.steps):
        trainer.step()
    torch.cuda.synchronize()
    if world > 1:
        torch.distributed.barrier()
    t1 = time.perf_counter()

    elapsed = t1 - t0
    if world > 1:
        t = torch.tensor([elapsed], device=device)
        torch.distributed.all_reduce(t, op=torch.distributed.ReduceOp.MAX)
        elapsed = t.item()

    total_images = world * args.batch_size * args.steps
    value = total_images / elapsed
    ms_per_step = elapsed / args.steps * 1000.0

    if rank == 0:
        result = {
            "metric": "train images/sec (whole node) at 112x112 bs=16/GPU",
            "value": value,
            "unit": "images/sec",
            "n_gpus": world,
            "steps": args.steps,
            "warmup": args.warmup,
            "ms_per_step": ms_per_step,
            "higher_is_better": True,
            "scaling": "weak",
            "vs_baseline": value / BASELINE_IMG_PER_SEC,
            "dtype": "bf16",
            "data": "synthetic",
            "config": {
                "model": "waternet",
                "global_batch": world * args.batch_size,
                "im_size": f"{args.height}x{args.width}",
                "parallelism": f"dp{world}",
                "loss": "0.05*vgg19_perceptual + mse255",
                "preprocess": "gpu wb/gamma/clahe in-step",
            },
        }
        print(json.dumps(result))

    if world > 1:
        torch.distributed.destroy_process_group()


if __name__ == "__main__":
    main()
