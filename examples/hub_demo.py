"""torch.hub usage demo — the reference's README quickstart
(/root/reference/README.md:44-64) against this repo:

    import torch
    preprocess, postprocess, model = torch.hub.load(
        "<repo-or-path>", "waternet", source="local",
        checkpoint="path/to/last.pt")   # or pretrained=True when online

Run: python examples/hub_demo.py [image.png]
"""

import sys
from pathlib import Path

import numpy as np
import torch

REPO = Path(__file__).resolve().parent.parent


def main():
    preprocess, postprocess, model = torch.hub.load(
        str(REPO), "waternet", source="local", pretrained=False,
        device="cuda:0" if torch.cuda.is_available() else "cpu",
    )
    if len(sys.argv) > 1:
        from PIL import Image

        rgb = np.asarray(Image.open(sys.argv[1]).convert("RGB"))
    else:
        rgb = np.random.default_rng(0).integers(
            0, 256, size=(112, 112, 3), dtype=np.uint8)

    dev = next(model.parameters()).device
    rgb_t, wb_t, he_t, gc_t = (t.to(dev) for t in preprocess(rgb))
    with torch.no_grad():
        out = model(rgb_t, wb_t, he_t, gc_t)
    enhanced = postprocess(out)[0]
    print(f"in {rgb.shape} -> out {enhanced.shape} {enhanced.dtype} "
          f"on {dev}")
    if len(sys.argv) > 1:
        from PIL import Image

        outp = Path(sys.argv[1]).with_suffix(".enhanced.png")
        Image.fromarray(enhanced).save(outp)
        print(f"wrote {outp}")


if __name__ == "__main__":
    main()
