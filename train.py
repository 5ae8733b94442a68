"""Training entry point — CLI-compatible with the reference train.py
(/root/reference/train.py:155-352): same flags (--epochs, --batch-size,
--height, --width, --weights, --seed), same savedir scheme (training/<n>),
same outputs (last.pt state_dict checkpoint each epoch, metrics-train.csv,
metrics-val.csv, config.json).

Engines:
  fast (default on a ROCm GPU) — the production step engine
    (waternet_amd.engine.fast.FastStepEngine): GPU preprocess inside the
    train step, native MFMA kernels end to end in NHWC bf16, fused Adam
    flat arena, whole-step hipGraph replay, on-device metric accumulation
    with ONE host sync per epoch. This is the same step bench.py measures —
    training through this CLI runs at benchmark throughput.
  eager — the PyTorch fp32 reference composition (CPU preprocess in the
    dataloader, torch.optim.Adam), matching the reference loop shape
    exactly (train.py:80-152). Used on CPU and for parity debugging.

Extensions (do not break reference invocations):
  --data-root      dataset root containing raw-890/ and reference-890/
  --synthetic N    use N synthetic image pairs instead of UIEB on disk
  --num-workers    DataLoader workers (reference default: 0)
  --shuffle        shuffle the train loader (reference leaves it False)
  --full-state     also save optimizer/scheduler/epoch sidecar for resume
  --engine         auto | fast | eager (auto = fast on GPU, eager on CPU)
  --no-graph       disable hipGraph capture in the fast engine

Multi-GPU: launch with `python -m torch.distributed.run --nproc-per-node N
train.py ...` — one rank per GPU, RCCL all-reduce gradient sync
(flat-arena collective inside the fast step / FlatBucketReducer in eager).
Rank 0 writes checkpoints/metrics.
"""

import argparse
import json
import os
from pathlib import Path
from timeit import default_timer as timer

import numpy as np
import torch

from waternet_amd.data.dataset import SyntheticUIEBDataset, UIEBDataset
from waternet_amd.engine.trainer import (
    TRAIN_METRICS_NAMES,
    VAL_METRICS_NAMES,
    eval_one_epoch,
    train_one_epoch,
)
from waternet_amd.models.waternet import WaterNet
from waternet_amd.parallel import (
    FlatBucketReducer,
    init_distributed,
    shard_dataset,
)


def parse_args(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--epochs", type=int, default=400)
    parser.add_argument("--batch-size", type=int, default=16)
    parser.add_argument("--height", type=int, default=112)
    parser.add_argument("--width", type=int, default=112)
    parser.add_argument("--weights", type=str, default=None,
                        help="(Optional) starting weights for training")
    parser.add_argument("--seed", type=int, default=None)
    parser.add_argument("--data-root", type=str, default="data")
    parser.add_argument("--synthetic", type=int, default=None, metavar="N",
                        help="use N synthetic image pairs instead of UIEB")
    parser.add_argument("--num-workers", type=int, default=None,
                        help="DataLoader workers; default 0 on the eager "
                             "engine (reference behavior) and 4 on the fast "
                             "engine (decode/collate hides under the GPU "
                             "step)")
    parser.add_argument("--shuffle", action="store_true")
    parser.add_argument("--full-state", action="store_true")
    parser.add_argument("--resume", type=str, default=None, metavar="DIR",
                        help="resume from a savedir holding last.pt (+ "
                             "last-trainstate.pt for optimizer/scheduler/"
                             "epoch state written by --full-state)")
    parser.add_argument("--engine", choices=["auto", "fast", "eager"],
                        default="auto",
                        help="auto = fast native engine on GPU, eager "
                             "PyTorch on CPU")
    parser.add_argument("--no-graph", action="store_true",
                        help="fast engine: disable hipGraph capture")
    return parser.parse_args(argv)


def _select_engine(args, device):
    if args.engine == "eager":
        return False
    fast_ok = (
        device.type == "cuda"
        and args.height % 8 == 0
        and args.width % 8 == 0
    )
    if fast_ok:
        from waternet_amd.ops import native_available

        fast_ok = native_available()
    if args.engine == "fast" and not fast_ok:
        from waternet_amd.ops import native_load_error

        raise SystemExit(
            "--engine fast requires a ROCm GPU with the native extension "
            f"and H,W divisible by 8 (load error: {native_load_error()})")
    return fast_ok


def _clean_state_dict(model):
    """Independent CPU tensors (the fast engine's params are views into the
    FusedAdam master arena; clone so last.pt matches the reference's
    standalone-tensor checkpoint layout, train.py:308)."""
    return {k: v.detach().cpu().clone()
            for k, v in model.state_dict().items()}


def _train_epoch_fast(engine, loader, batch_size, device, epoch_num,
                      total_epochs, progress):
    snap = engine.metrics_snapshot()
    t0 = timer()
    iterator = loader
    if progress:
        try:
            from tqdm import tqdm

            iterator = tqdm(loader, total=len(loader), ascii=True,
                            desc=f"Epoch {epoch_num + 1}/{total_epochs}",
                            bar_format="{l_bar}{bar:20}{r_bar}")
        except ImportError:
            pass
    n_images = 0
    for batch in iterator:
        raw, ref = batch["raw"], batch["ref"]
        n_images += raw.size(0)
        if raw.size(0) == batch_size:
            engine.load_batch(raw, ref)
            engine.step()
        else:  # ragged tail batch: eager step at its true size
            engine.step_batch(raw.to(device, non_blocking=True),
                              ref.to(device, non_blocking=True))
    m = engine.metrics_since(snap)  # ONE host sync per epoch (joins the GPU)
    wall = timer() - t0
    if progress and n_images:
        # wall includes queueing + the GPU completing every queued step
        # (metrics_since syncs); this is the per-rank TRAIN throughput —
        # the number to compare against bench.py
        print(f"    [fast] epoch train wall {wall:.3f}s = "
              f"{n_images / wall:.0f} img/s")
    return {
        "mse": m["mse255"], "ssim": m["ssim"], "psnr": m["psnr"],
        "perceptual_loss": m["perceptual"], "loss": m["loss"],
    }


def _eval_epoch_fast(engine, loader, device):
    engine.reset_eval()
    for batch in loader:
        engine.eval_batch(batch["raw"].to(device, non_blocking=True),
                          batch["ref"].to(device, non_blocking=True))
    return engine.eval_metrics()


def main(argv=None):
    start_ts = timer()
    args = parse_args(argv)

    dist_env = init_distributed()
    rank0 = dist_env.rank == 0

    projectroot = Path(__file__).parent
    # savedir root: training/ beside train.py (reference behavior), or
    # $WATERNET_TRAINING_DIR (tests / packaged installs)
    outputdir = Path(os.environ.get("WATERNET_TRAINING_DIR",
                                    projectroot / "training"))
    torch.manual_seed(0)  # always, as the reference (train.py:160)
    if args.seed is not None:
        torch.manual_seed(args.seed)

    if torch.cuda.is_available():
        torch.cuda.set_device(dist_env.local_rank)
        device = torch.device("cuda", dist_env.local_rank)
    else:
        device = torch.device("cpu")

    use_fast = _select_engine(args, device)

    # Savedir: training/<max numbered subdir + 1>  (train.py:210-221)
    outputdir.mkdir(exist_ok=True)
    nums = [int(p.stem) for p in outputdir.glob("*")
            if p.is_dir() and p.stem.isdecimal()]
    savedir = outputdir / str(max(nums) + 1 if nums else 0)

    # Dataset + split (800/90 on UIEB; proportional for synthetic).
    # Fast engine: datasets yield uint8 HWC pairs; wb/gamma/clahe run
    # on-GPU inside the step (the reference's CPU transform loop is its
    # throughput bottleneck — SURVEY §6).
    if args.synthetic is not None:
        dataset = SyntheticUIEBDataset(
            n_images=args.synthetic, im_height=args.height,
            im_width=args.width, raw_mode=use_fast,
        )
        n_val = max(1, int(0.1 * len(dataset)))
        split = [len(dataset) - n_val, n_val]
    else:
        dataset = UIEBDataset(
            Path(args.data_root) / "raw-890",
            Path(args.data_root) / "reference-890",
            im_height=args.height,
            im_width=args.width,
            raw_mode=use_fast,
        )
        split = [800, 90]
    if len(dataset) != sum(split):
        raise SystemExit(
            f"the UIEB 800/90 split (reference train.py:233) needs exactly "
            f"{sum(split)} images; found {len(dataset)} under "
            f"{args.data_root} (use --synthetic N for on-the-fly data)")
    train_dataset, val_dataset = torch.utils.data.random_split(dataset, split)
    train_dataset = shard_dataset(train_dataset, dist_env)

    num_workers = args.num_workers
    if num_workers is None:
        num_workers = min(4, os.cpu_count() or 1) if use_fast else 0

    train_loader = torch.utils.data.DataLoader(
        train_dataset, batch_size=args.batch_size, shuffle=args.shuffle,
        num_workers=num_workers, pin_memory=device.type == "cuda",
        persistent_workers=num_workers > 0,
    )
    val_loader = torch.utils.data.DataLoader(
        val_dataset, batch_size=args.batch_size,
        num_workers=num_workers,
        pin_memory=device.type == "cuda",
        persistent_workers=num_workers > 0,
    )

    if rank0:
        print(f"Using device: {device} "
              f"({'fast native engine' if use_fast else 'eager engine'})")

    model = WaterNet()
    if args.resume is not None:
        ckpt = Path(args.resume) / "last.pt"
        if not ckpt.is_file():
            raise SystemExit(
                f"--resume {args.resume}: no last.pt there — pass the "
                "savedir of a previous run (e.g. training/0)")
        model.load_state_dict(torch.load(ckpt, map_location="cpu"))
    elif args.weights is not None:
        with open(args.weights, "rb") as f:
            model.load_state_dict(torch.load(f, map_location="cpu"))
    model.to(device)
    model.train()

    engine = None
    reducer = None
    if use_fast:
        from waternet_amd.engine.fast import FastStepEngine

        engine = FastStepEngine(
            model, batch_size=args.batch_size, height=args.height,
            width=args.width, device=device,
            world_size=dist_env.world_size,
            # keep the RCCL collective out of graph capture under DDP
            # (same policy as bench.py)
            use_graph=(not args.no_graph) and dist_env.world_size == 1,
        )
        optimizer, scheduler = engine.opt, engine.sched
        vgg_model = engine.vgg
    else:
        from waternet_amd.models.vgg import PerceptualModel

        optimizer = torch.optim.Adam(model.parameters(), lr=1e-3)
        scheduler = torch.optim.lr_scheduler.StepLR(
            optimizer, step_size=10000, gamma=0.1
        )
        vgg_model = PerceptualModel().to(device).eval()

    start_epoch = 0
    if args.resume is not None:
        # full resume: optimizer/scheduler/epoch from the sidecar if present
        # (the reference's resume via --weights restarts the LR schedule —
        # SURVEY §5.3; the sidecar fixes that without changing last.pt)
        sidecar = Path(args.resume) / "last-trainstate.pt"
        if sidecar.exists():
            state = torch.load(sidecar, map_location="cpu")
            opt_state = state.get("optimizer", {})
            # FusedAdam stores its moments under "wn_fused"; a sidecar
            # written by the eager engine (torch.optim.Adam) has per-param
            # states FusedAdam cannot consume — loading it would silently
            # reset the moments, so say so and skip the optimizer.
            cross_engine = use_fast != ("wn_fused" in opt_state)
            try:
                if cross_engine:
                    if rank0:
                        print("WARNING: sidecar optimizer state was written "
                              "by the other engine (fast<->eager); optimizer "
                              "moments reset, LR schedule/epoch restored")
                else:
                    optimizer.load_state_dict(state["optimizer"])
                scheduler.load_state_dict(state["scheduler"])
                start_epoch = state["epoch"] + 1
                if rank0:
                    print(f"Resumed epoch {start_epoch} from {sidecar}")
            except (KeyError, ValueError) as e:
                if rank0:
                    print(f"WARNING: sidecar optimizer state incompatible "
                          f"with this engine ({e!r}); optimizer state reset")

    if dist_env.world_size > 1:
        if use_fast:
            # all ranks start from rank0's flat master arena (one-time)
            torch.distributed.broadcast(engine.opt.master, src=0)
        else:
            # Broadcast initial params so all ranks start identical, then
            # all-reduce gradients each step through one flat bucket.
            reducer = FlatBucketReducer(model, dist_env)
            reducer.broadcast_params()

    saved_train = {k: [] for k in TRAIN_METRICS_NAMES}
    saved_val = {k: [] for k in VAL_METRICS_NAMES}

    for epoch in range(start_epoch, args.epochs):
        if use_fast:
            train_metrics = _train_epoch_fast(
                engine, train_loader, args.batch_size, device,
                epoch_num=epoch, total_epochs=args.epochs, progress=rank0,
            )
            val_metrics = _eval_epoch_fast(engine, val_loader, device)
        else:
            train_metrics = train_one_epoch(
                model, train_loader, optimizer, scheduler, vgg_model, device,
                epoch_num=epoch, total_epochs=args.epochs,
                grad_reducer=reducer, progress=rank0,
            )
            val_metrics = eval_one_epoch(model, val_loader, device, vgg_model)

        if dist_env.world_size > 1:
            train_metrics = dist_env.average_metrics(train_metrics)
            val_metrics = dist_env.average_metrics(val_metrics)

        if rank0:
            print("    Train ||",
                  "   ".join(f"{k}: {v:.03g}" for k, v in train_metrics.items()))
            print("    Val   ||",
                  "   ".join(f"{k}: {v:.03g}" for k, v in val_metrics.items()))
            print()

        for k, v in train_metrics.items():
            saved_train[k].append(v)
        for k, v in val_metrics.items():
            saved_val[k].append(v)

        if rank0:
            savedir.mkdir(exist_ok=True)
            torch.save(_clean_state_dict(model), savedir / "last.pt")
            if args.full_state:
                torch.save(
                    {
                        "epoch": epoch,
                        "optimizer": optimizer.state_dict(),
                        "scheduler": scheduler.state_dict(),
                    },
                    savedir / "last-trainstate.pt",
                )

    if not saved_train[TRAIN_METRICS_NAMES[0]]:
        # zero epochs ran (e.g. --resume of an already-finished run, or
        # --epochs 0): nothing to save — avoid np.stack([]) crashing
        if rank0:
            print(f"No epochs to run (start epoch {start_epoch} >= "
                  f"{args.epochs}); nothing saved")
        return

    if rank0:
        savedir.mkdir(exist_ok=True)
        train_arr = np.stack(
            [np.asarray(saved_train[k]) for k in TRAIN_METRICS_NAMES], axis=1
        )
        val_arr = np.stack(
            [np.asarray(saved_val[k]) for k in VAL_METRICS_NAMES], axis=1
        )
        np.savetxt(savedir / "metrics-train.csv", train_arr, fmt="%f",
                   delimiter=",", comments="",
                   header=",".join(TRAIN_METRICS_NAMES))
        np.savetxt(savedir / "metrics-val.csv", val_arr, fmt="%f",
                   delimiter=",", comments="",
                   header=",".join(VAL_METRICS_NAMES))
        with open(savedir / "config.json", "w") as f:
            json.dump(
                {
                    "epochs": args.epochs,
                    "batch_size": args.batch_size,
                    "im_height": args.height,
                    "im_width": args.width,
                    "weights": args.weights,
                },
                f,
                indent=4,
            )
        print(f"Metrics and weights saved to {savedir}")
        print(f"Total time: {timer() - start_ts}s")


if __name__ == "__main__":
    main()
