"""End-to-end quickstart — the analog of the reference's Colab example
(/root/reference/colab-example-waternet.ipynb): train a few epochs, score
the checkpoint, enhance an image, all on synthetic data (no downloads).

    python examples/quickstart.py            # CPU ok; fast engine on GPU

On a GPU this exercises the full native stack: GPU preprocess in-step,
MFMA convolutions, hipGraph train step, native eval, and the
hipGraph-captured per-frame inference engine.
"""

import os
import sys
import tempfile
from pathlib import Path

import numpy as np
import torch

REPO = Path(__file__).resolve().parent.parent
sys.path.insert(0, str(REPO))


def main():
    import score as score_cli
    import train as train_cli

    workdir = Path(tempfile.mkdtemp(prefix="waternet_quickstart_"))
    os.environ["WATERNET_TRAINING_DIR"] = str(workdir)
    on_gpu = torch.cuda.is_available()
    size = "64" if not on_gpu else "112"
    epochs = "1" if not on_gpu else "5"

    print(f"== train ({'GPU fast engine' if on_gpu else 'CPU eager'}) ==")
    train_cli.main(["--synthetic", "64", "--epochs", epochs,
                    "--batch-size", "8", "--height", size, "--width", size,
                    "--full-state"])
    ckpt = workdir / "0" / "last.pt"

    print("\n== score ==")
    score_cli.main(["--weights", str(ckpt), "--synthetic", "64",
                    "--batch-size", "8", "--height", size, "--width", size])

    print("\n== enhance one frame ==")
    from waternet_amd import WaterNet

    model = WaterNet()
    model.load_state_dict(torch.load(ckpt, map_location="cpu"))
    rng = np.random.default_rng(0)
    frame = rng.integers(0, 256, size=(int(size), int(size), 3),
                         dtype=np.uint8)
    if on_gpu:
        from waternet_amd.engine.inferencer import InferenceEngine

        eng = InferenceEngine(model.to("cuda:0"), int(size), int(size))
        out = eng.infer_frame(frame)
    else:
        from waternet_amd.data.bridge import arr2ten, ten2arr
        from waternet_amd import transform

        wb, gc, he = transform(frame)
        with torch.no_grad():
            t = model(arr2ten(frame, True), arr2ten(wb, True),
                      arr2ten(he, True), arr2ten(gc, True))
        out = ten2arr(t)[0]
    print(f"enhanced frame: {out.shape} {out.dtype}; artifacts in {workdir}")


if __name__ == "__main__":
    main()
