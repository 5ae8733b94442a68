"""Minimal production serving endpoint for the WaterNet engine.

    python serve.py --host 0.0.0.0 --port 8000 --weights training/0/last.pt

POST /enhance with a PNG/JPEG body (or multipart file) returns the enhanced
image as PNG. GET /healthz reports device + model status. On a ROCm GPU the
per-request pipeline is the hipGraph-captured InferenceEngine (fixed-shape
graphs cached per resolution); on CPU it falls back to the eager path.
"""

import argparse
import io
from collections import OrderedDict
from typing import Optional

import numpy as np
import torch

from waternet_amd.models.waternet import WaterNet

# A graph-captured engine holds static buffers + activation workspace per
# resolution; bound the cache so clients posting many distinct resolutions
# cannot exhaust HBM (oldest engine evicted, its graph memory freed).
MAX_ENGINES = 8
MAX_DIM = 4096  # reject images beyond 4096 px per side (~50 MB fp32 @4k)
MAX_BODY_BYTES = 64 * 1024 * 1024


class Server:
    def __init__(self, weights: Optional[str] = None, device: Optional[str] = None):
        self.device = torch.device(
            device or ("cuda:0" if torch.cuda.is_available() else "cpu")
        )
        torch.manual_seed(0)
        self.model = WaterNet()
        if weights:
            self.model.load_state_dict(torch.load(weights, map_location="cpu"))
        self.model.to(self.device).eval()
        self._engines = OrderedDict()  # (H, W) -> InferenceEngine LRU

    def enhance(self, rgb_u8: np.ndarray) -> np.ndarray:
        h, w = rgb_u8.shape[:2]
        if h > MAX_DIM or w > MAX_DIM:
            raise ValueError(
                f"image {w}x{h} exceeds the {MAX_DIM}px per-side limit")
        if self.device.type == "cuda":
            from waternet_amd.engine.inferencer import InferenceEngine, pad8

            # any resolution runs on GPU: reflect-pad to /8 (the CLAHE
            # tile grid), crop back after — also coalesces the engine
            # cache onto /8 keys
            padded, oh, ow = pad8(rgb_u8)
            ph, pw = padded.shape[:2]
            eng = self._engines.get((ph, pw))
            if eng is None:
                eng = InferenceEngine(self.model, ph, pw, device=self.device)
                while len(self._engines) >= MAX_ENGINES:
                    self._engines.popitem(last=False)
                self._engines[(ph, pw)] = eng
            else:
                self._engines.move_to_end((ph, pw))
            return eng.infer_frame(padded)[:oh, :ow]
        # CPU fallback: reference transforms + eager forward
        from waternet_amd.data.bridge import arr2ten, ten2arr
        from waternet_amd.data.transforms import transform

        wb, gc, he = transform(rgb_u8)
        with torch.no_grad():
            out = self.model(
                arr2ten(rgb_u8, True).to(self.device),
                arr2ten(wb, True).to(self.device),
                arr2ten(he, True).to(self.device),
                arr2ten(gc, True).to(self.device),
            )
        return ten2arr(out)[0]


def create_app(weights: Optional[str] = None, device: Optional[str] = None):
    from fastapi import FastAPI, Request, Response

    from PIL import Image

    app = FastAPI(title="waternet-mi355x")
    server = Server(weights=weights, device=device)

    @app.get("/healthz")
    def healthz():
        return {
            "status": "ok",
            "device": str(server.device),
            "engines": list(map(list, server._engines.keys())),
        }

    @app.post("/enhance")
    async def enhance(request: Request):
        body = await request.body()
        if len(body) > MAX_BODY_BYTES:
            return Response(status_code=413, content="image body too large")
        try:
            # PIL's MAX_IMAGE_PIXELS decompression-bomb guard applies here;
            # an oversized or undecodable body is a client error, not a 500
            img = Image.open(io.BytesIO(body)).convert("RGB")
        except Exception as e:  # noqa: BLE001
            return Response(status_code=400,
                            content=f"body is not a decodable image: {e}")
        rgb = np.asarray(img)
        try:
            out = server.enhance(rgb)
        except ValueError as e:
            return Response(status_code=413, content=str(e))
        buf = io.BytesIO()
        Image.fromarray(out).save(buf, format="PNG")
        return Response(content=buf.getvalue(), media_type="image/png")

    return app


def main(argv=None):
    parser = argparse.ArgumentParser()
    parser.add_argument("--host", default="127.0.0.1")
    parser.add_argument("--port", type=int, default=8000)
    parser.add_argument("--weights", default=None)
    parser.add_argument("--device", default=None)
    args = parser.parse_args(argv)

    import uvicorn

    uvicorn.run(create_app(args.weights, args.device), host=args.host,
                port=args.port)


if __name__ == "__main__":
    main()
