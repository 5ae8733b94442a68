"""Inference entry point — CLI-compatible with the reference inference.py
(/root/reference/inference.py): --source image/video/directory, --weights,
--name, --show-split. Images are enhanced one by one; videos frame by frame
(bs=1, fixed shapes — on GPU this path is hipGraph-captured via
waternet_amd.engine.inferencer).

Differences from the reference, forced by this environment:
  - PIL replaces OpenCV for image IO (no cv2 wheel offline); --show-split
    text is drawn with PIL instead of cv2.putText.
  - Video IO uses the ffmpeg binary when present; otherwise video sources
    raise a clear error (no cv2.VideoCapture available).
  - No weight auto-download (no network): --weights is optional but
    random-init weights produce garbage, so a warning is printed.
"""

import argparse
import sys
from pathlib import Path

import numpy as np
import torch

from waternet_amd.data.bridge import arr2ten, ten2arr
from waternet_amd.data.transforms import transform
from waternet_amd.models.waternet import WaterNet

# strict supersets of the reference's lists (inference.py:17-18: bmp/jpg/
# jpeg/png/gif and mp4/mpeg/avi) — PIL takes the first frame of a .gif
IM_SUFFIXES = [".bmp", ".jpg", ".jpeg", ".png", ".gif", ".tiff", ".webp"]
VID_SUFFIXES = [".mp4", ".mpeg", ".avi", ".mov", ".mkv", ".webm"]


def parse_args(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--source", type=str, required=True,
                        help="image / video / directory of images")
    parser.add_argument("--weights", type=str, default=None)
    parser.add_argument("--name", type=str, default=None,
                        help="output subdirectory name under ./output")
    parser.add_argument("--show-split", action="store_true",
                        help="write before/after composites")
    return parser.parse_args(argv)


def make_savedir(name):
    """Pick output/<name or next number>. The directory itself is created
    as late as possible — at the first write — so failed runs leave no
    empty savedirs (reference inference.py:198-200)."""
    outputdir = Path("output")
    if name is not None:
        return outputdir / name
    nums = [int(p.stem) for p in outputdir.glob("*")
            if p.is_dir() and p.stem.isdecimal()] if outputdir.is_dir() \
        else []
    return outputdir / str(max(nums) + 1 if nums else 0)


def load_model(weights, device):
    """--weights path, or the reference's auto-download scheme (reference
    inference.py:15-21,103-109): fetch the published hash-named checkpoint
    next to this script with torch.hub's check_hash; falls back to a
    random-init warning when offline."""
    model = WaterNet()
    if weights is not None:
        with open(weights, "rb") as f:
            model.load_state_dict(torch.load(f, map_location="cpu"))
    else:
        import hubconf

        wd = Path(__file__).resolve().parent
        local = wd / hubconf.WEIGHTS_FILE
        try:
            if local.exists():
                model.load_state_dict(torch.load(local, map_location="cpu"))
            else:
                sd = torch.hub.load_state_dict_from_url(
                    hubconf.WEIGHTS_URL, map_location="cpu",
                    model_dir=str(wd), file_name=hubconf.WEIGHTS_FILE,
                    check_hash=True,
                )
                model.load_state_dict(sd)
        except Exception as e:  # noqa: BLE001 — offline environment
            print(
                "WARNING: no --weights given and auto-download failed "
                f"({e!r}); using random-init weights.",
                file=sys.stderr,
            )
    model.to(device).eval()
    return model


_ENGINES = {}


def _gpu_engine(model, h, w, device):
    key = (id(model), h, w)
    eng = _ENGINES.get(key)
    if eng is None:
        from waternet_amd.engine.inferencer import InferenceEngine

        eng = InferenceEngine(model, h, w, device=device)
        _ENGINES[key] = eng
    return eng


@torch.no_grad()
def enhance_frame(model, rgb: np.ndarray, device) -> np.ndarray:
    """uint8 HWC RGB -> enhanced uint8 HWC RGB. On GPU the whole frame
    pipeline (preprocess + forward + postprocess) runs as a hipGraph-
    captured on-device pipeline; on CPU the reference-semantics numpy
    transforms + eager model run."""
    if device.type == "cuda":
        from waternet_amd.ops import native_available

        if native_available():
            from waternet_amd.engine.inferencer import pad8

            padded, h, w = pad8(rgb)
            eng = _gpu_engine(model, padded.shape[0], padded.shape[1],
                              device)
            return eng.infer_frame(padded)[:h, :w]
    wb, gc, he = transform(rgb)
    rgb_ten = arr2ten(rgb, add_batch_dim=True).to(device)
    wb_ten = arr2ten(wb, add_batch_dim=True).to(device)
    gc_ten = arr2ten(gc, add_batch_dim=True).to(device)
    he_ten = arr2ten(he, add_batch_dim=True).to(device)
    # he fills the `ce` slot — reference call order (inference.py:191)
    out = model(rgb_ten, wb_ten, he_ten, gc_ten)
    return ten2arr(out)[0]


def compose_split(before: np.ndarray, after: np.ndarray) -> np.ndarray:
    """Left half original ('Before'), right half enhanced ('After'),
    replicating the reference's --show-split composite
    (inference.py:202-233)."""
    from PIL import Image, ImageDraw

    h, w = before.shape[:2]
    comp = np.concatenate([before[:, : w // 2], after[:, w // 2:]], axis=1)
    im = Image.fromarray(comp)
    draw = ImageDraw.Draw(im)
    draw.line([(w // 2, 0), (w // 2, h)], fill=(255, 255, 255), width=2)
    draw.text((10, 10), "Before", fill=(255, 255, 255))
    draw.text((w // 2 + 10, 10), "After", fill=(255, 255, 255))
    return np.asarray(im)


def run_image(model, path: Path, savedir: Path, device, show_split):
    from PIL import Image

    with Image.open(path) as im:
        rgb = np.asarray(im.convert("RGB"))
    out = enhance_frame(model, rgb, device)
    result = compose_split(rgb, out) if show_split else out
    savedir.mkdir(parents=True, exist_ok=True)  # late, ref inference.py:198
    Image.fromarray(result).save(savedir / path.name)


def run_video(model, path: Path, savedir: Path, device, show_split):
    from waternet_amd.engine.video import FFmpegReader, FFmpegWriter

    reader = FFmpegReader(path)
    savedir.mkdir(parents=True, exist_ok=True)  # late, ref inference.py:198
    outpath = savedir / path.name
    writer = FFmpegWriter(outpath, reader.width, reader.height, reader.fps)
    n = 0
    for frame in reader:
        out = enhance_frame(model, frame, device)
        writer.write(compose_split(frame, out) if show_split else out)
        n += 1
        if n % 50 == 0:
            print(f"{n} frames processed")
    writer.close()
    print(f"Wrote {n} frames to {outpath}")


def main(argv=None):
    args = parse_args(argv)
    device = torch.device("cuda" if torch.cuda.is_available() else "cpu")
    print(f"Using device: {device}")

    source = Path(args.source)
    if not source.exists():
        raise SystemExit(f"{args.source} does not exist!")
    if source.is_dir():
        # both images and videos, as the reference (inference.py:128-135)
        sources = sorted(
            p for p in source.iterdir()
            if p.suffix.lower() in IM_SUFFIXES
            or p.suffix.lower() in VID_SUFFIXES
        )
    else:
        sources = [source]
    print(f"Total images/videos: {len(sources)}")

    model = load_model(args.weights, device)
    savedir = make_savedir(args.name)

    for p in sources:
        if p.suffix.lower() in VID_SUFFIXES:
            run_video(model, p, savedir, device, args.show_split)
        elif p.suffix.lower() in IM_SUFFIXES:
            run_image(model, p, savedir, device, args.show_split)
        else:
            print(f"Skipping unrecognized suffix: {p}", file=sys.stderr)

    print(f"Results saved to {savedir}")


if __name__ == "__main__":
    main()
