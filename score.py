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
    parser.add_argument("--batch-size", type=int, default=16)
    parser.add_argument("--height", type=int, default=112)
    parser.add_argument("--width", type=int, default=112)
    parser.add_argument("--seed", type=int, default=None)
    parser.add_argument("--data-root", type=str, default="data")
    parser.add_argument("--synthetic", type=int, default=None)
    return parser.parse_args(argv)


def main(argv=None):
    args = parse_args(argv)
    torch.manual_seed(0)
    if args.seed is not None:
        torch.manual_seed(args.seed)

    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")

    if args.synthetic is not None:
        dataset = SyntheticUIEBDataset(
            n_images=args.synthetic, im_height=args.height, im_width=args.width
        )
        n_val = max(1, int(0.1 * len(dataset)))
        split = [len(dataset) - n_val, n_val]
    else:
        dataset = UIEBDataset(
            Path(args.data_root) / "raw-890",
            Path(args.data_root) / "reference-890",
            im_height=args.height,
            im_width=args.width,
        )
        split = [800, 90]
    _, val_dataset = torch.utils.data.random_split(dataset, split)
    val_loader = torch.utils.data.DataLoader(val_dataset,
                                             batch_size=args.batch_size)

    model = WaterNet()
    with open(args.weights, "rb") as f:
        model.load_state_dict(torch.load(f, map_location="cpu"))
    model.to(device).eval()

    vgg_model = PerceptualModel().to(device).eval()

    metrics = eval_one_epoch(model, val_loader, device, vgg_model)
    pprint(metrics)
    return metrics


if __name__ == "__main__":
    main()
