"""Validation scoring entry point — CLI-compatible with the reference
score.py (/root/reference/score.py:84-177): rebuilds the seed-dependent
train/val split, loads --weights, runs eval over the val split, prints the
metrics dict.
"""

import argparse
from pathlib import Path
from pprint import pprint

import torch

from waternet_amd.data.dataset import SyntheticUIEBDataset, UIEBDataset
from waternet_amd.engine.trainer import eval_one_epoch
from waternet_amd.models.vgg import PerceptualModel
from waternet_amd.models.waternet import WaterNet


def parse_args(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--weights", type=str, required=True)
    # accepted-but-unused, as the reference (score.py:99-101: scoring takes
    # no epochs; the flag exists there and must keep parsing here)
    parser.add_argument("--epochs", type=int, default=400,
                        help="(unused; reference CLI compatibility)")
    parser.add_argument("--batch-size", type=int, default=16)
    parser.add_argument("--height", type=int, default=112)
    parser.add_argument("--width", type=int, default=112)
    parser.add_argument("--seed", type=int, default=None)
    parser.add_argument("--data-root", type=str, default="data")
    parser.add_argument("--synthetic", type=int, default=None)
    parser.add_argument("--engine", choices=["auto", "fast", "eager"],
                        default="auto",
                        help="fast = GPU preprocess + native eval (same "
                             "metrics; minutes -> seconds on real datasets)")
    parser.add_argument("--num-workers", type=int, default=0,
                        help="DataLoader workers for the val loader "
                             "(parallelizes the CPU preprocess on the "
                             "eager engine; reference default 0)")
    return parser.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    torch.manual_seed(0)
    if args.seed is not None:
        torch.manual_seed(args.seed)

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    use_fast = args.engine != "eager" and device.type == "cuda" \
        and args.height % 8 == 0 and args.width % 8 == 0
    if use_fast:
        from waternet_amd.ops import native_available

        use_fast = native_available()
    if args.engine == "fast" and not use_fast:
        raise SystemExit("--engine fast requires a ROCm GPU with the "
                         "native extension and H,W divisible by 8")

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
            f"{args.data_root}")
    _, val_dataset = torch.utils.data.random_split(dataset, split)
    val_loader = torch.utils.data.DataLoader(val_dataset,
                                             batch_size=args.batch_size,
                                             num_workers=args.num_workers)

    model = WaterNet()
    if not Path(args.weights).is_file():
        raise SystemExit(f"weights file not found: {args.weights}")
    with open(args.weights, "rb") as f:
        model.load_state_dict(torch.load(f, map_location="cpu"))
    model.to(device).eval()

    vgg_model = PerceptualModel().to(device).eval()

    if use_fast:
        # GPU preprocess + native NHWC eval (same metric definitions; the
        # same-checkpoint eager/fast scores cross-validated within noise)
        from waternet_amd.engine.fast import VAL_KEYS, eval_metrics_batch

        sums = torch.zeros(4, dtype=torch.float64, device=device)
        n_batches = 0
        for batch in val_loader:
            sums += eval_metrics_batch(
                model, vgg_model,
                batch["raw"].to(device, non_blocking=True),
                batch["ref"].to(device, non_blocking=True))
            n_batches += 1
        vals = (sums / max(n_batches, 1)).tolist()
        metrics = dict(zip(VAL_KEYS, vals))
    else:
        metrics = eval_one_epoch(model, val_loader, device, vgg_model)
    pprint(metrics)
    return metrics


if __name__ == "__main__":
    main()
