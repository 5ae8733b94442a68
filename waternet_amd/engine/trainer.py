"""Epoch-level training / evaluation loops.

Mirrors the reference loops (train_one_epoch train.py:80-152,
eval_one_epoch train.py:26-77) with two documented fixes:
  - eval perceptual_loss is accumulated with += (the reference assigns `=`
    so its reported value is last-batch/num_batches — train.py:71,74).
  - the VGG model is an explicit argument to eval (the reference reads a
    module-level global — train.py:59).

Under distributed data parallelism each rank runs its shard and metrics
are all-reduce averaged.
"""

import torch

from waternet_amd.engine.losses import composite_loss, perceptual_loss, mse_loss
from waternet_amd.utils.metrics import (
    peak_signal_noise_ratio,
    structural_similarity_index_measure,
)

TRAIN_METRICS_NAMES = ["mse", "ssim", "psnr", "perceptual_loss", "loss"]
VAL_METRICS_NAMES = ["mse", "ssim", "psnr", "perceptual_loss"]


def _move_batch(batch, device):
    return {k: v.to(device, non_blocking=True) for k, v in batch.items()}


def train_one_epoch(
    model,
    train_dataloader,
    optimizer,
    scheduler,
    vgg_model,
    device,
    epoch_num=0,
    total_epochs=1,
    grad_reducer=None,
    progress=True,
):
    """One epoch of training. grad_reducer: optional callable invoked after
    backward and before optimizer.step() (DDP flat-bucket all-reduce)."""
    model.train()
    epoch_metrics = {k: 0.0 for k in TRAIN_METRICS_NAMES}
    n_batches = len(train_dataloader)

    iterator = train_dataloader
    if progress:
        try:
            from tqdm import tqdm

            iterator = tqdm(
                train_dataloader,
                total=n_batches,
                ascii=True,
                desc=f"Epoch {epoch_num + 1}/{total_epochs}",
                bar_format="{l_bar}{bar:20}{r_bar}",
            )
        except ImportError:
            pass

    for idx, batch in enumerate(iterator):
        batch = _move_batch(batch, device)
        # NOTE argument order: the he tensor feeds the `ce` slot, matching
        # the reference call (train.py:108 -> net.py:99).
        out = model(batch["raw"], batch["wb"], batch["he"], batch["gc"])

        loss, ploss, mloss = composite_loss(out, batch["ref"], vgg_model)

        optimizer.zero_grad(set_to_none=True)
        loss.backward()
        if grad_reducer is not None:
            grad_reducer()
        optimizer.step()
        if scheduler is not None:
            scheduler.step()  # per-minibatch, as the reference (train.py:133)

        epoch_metrics["loss"] += loss.item()
        epoch_metrics["perceptual_loss"] += ploss.item()
        epoch_metrics["mse"] += mloss.item()
        with torch.no_grad():
            ssim = structural_similarity_index_measure(
                preds=out, target=batch["ref"]
            )
            psnr = peak_signal_noise_ratio(
                preds=out, target=batch["ref"], data_range=1.0
            )
        epoch_metrics["ssim"] += ssim.item()
        epoch_metrics["psnr"] += psnr.item()

        if progress and hasattr(iterator, "set_postfix") and idx % 10 == 0 and idx:
            iterator.set_postfix({"loss": loss.item()})

    return {k: v / max(n_batches, 1) for k, v in epoch_metrics.items()}


@torch.no_grad()
def eval_one_epoch(model, val_dataloader, device, vgg_model):
    import os

    # Reference mode (WATERNET_REFERENCE_EVAL_BUG=1): replicate the
    # reference's `=` instead of `+=` at train.py:71, which makes the
    # reported val perceptual = last-batch value / n_batches — needed only
    # when parity-scoring against the reference's own printed numbers
    # (SURVEY §7 behavioral quirks).
    replicate_bug = os.environ.get("WATERNET_REFERENCE_EVAL_BUG", "0") == "1"
    model.eval()
    epoch_metrics = {k: 0.0 for k in VAL_METRICS_NAMES}
    n_batches = len(val_dataloader)

    for batch in val_dataloader:
        batch = _move_batch(batch, device)
        out = model(batch["raw"], batch["wb"], batch["he"], batch["gc"])

        ploss = perceptual_loss(out, batch["ref"], vgg_model)
        if replicate_bug:
            epoch_metrics["perceptual_loss"] = ploss.item()
        else:
            epoch_metrics["perceptual_loss"] += ploss.item()
        epoch_metrics["mse"] += mse_loss(out, batch["ref"]).item()
        epoch_metrics["ssim"] += structural_similarity_index_measure(
            preds=out, target=batch["ref"]
        ).item()
        epoch_metrics["psnr"] += peak_signal_noise_ratio(
            preds=out, target=batch["ref"], data_range=1.0
        ).item()

    model.train()
    return {k: v / max(n_batches, 1) for k, v in epoch_metrics.items()}
