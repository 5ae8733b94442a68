"""hipGraph-captured per-frame inference engine (BASELINE config 4).

The reference's video loop (inference.py:261-323) runs CPU preprocess +
bs=1 forward + CPU postprocess per frame. This engine keeps the whole frame
pipeline on-GPU with fixed shapes and captures it in ONE hipGraph:

  raw u8 frame (static buffer) -> GPU preprocess (wb/gamma/clahe)
  -> fused input build -> WaterNet forward (MFMA kernels)
  -> fused postprocess (clip*255 -> u8)

Per frame at steady state: one H2D copy in, one graph replay, one D2H copy
out. Falls back to eager kernel launches if capture fails.
"""

import numpy as np
import torch

from waternet_amd.ops import ext
from waternet_amd.ops.preprocess import gpu_transform_batch


def pad8(rgb_u8: np.ndarray):
    """Reflect-pad an HWC frame to the next multiple of 8 per side (the
    GPU CLAHE tile-grid requirement). Returns (padded, orig_h, orig_w).

    Note: the padded rows/cols (<= 7 px, reflected) participate in the
    WB quantile and CLAHE tile statistics — a boundary-only deviation
    from the CPU transforms, accepted so ANY resolution runs the
    hipGraph GPU pipeline (crop back with [:h, :w] after)."""
    h, w = rgb_u8.shape[:2]
    ph, pw = (-h) % 8, (-w) % 8
    if ph == 0 and pw == 0:
        return rgb_u8, h, w
    return (np.pad(rgb_u8, ((0, ph), (0, pw), (0, 0)), mode="reflect"),
            h, w)


class InferenceEngine:
    def __init__(self, model, height, width, device="cuda:0",
                 use_graph=True):
        self.model = model.eval()
        self.device = torch.device(device)
        self.h, self.w = height, width
        self.raw_static = torch.empty(1, height, width, 3,
                                      dtype=torch.uint8, device=self.device)
        self.out_static = None
        self._graph = None
        self._use_graph = use_graph and height % 8 == 0 and width % 8 == 0

    @torch.no_grad()
    def _body(self):
        from waternet_amd.engine.native import waternet_forward_from_inputs

        raw = self.raw_static
        wb, gc, he = gpu_transform_batch(raw)
        e = ext()
        # he fills the ce slot (reference inference.py:191 -> net.py:99);
        # full-NHWC: fused uint8 cat-fold -> convs -> fused u8 postprocess,
        # no NCHW tensor anywhere in the frame pipeline.
        inputs = e.build_inputs_u8(raw, wb, he, gc)
        out_nhwc = waternet_forward_from_inputs(self.model, *inputs)
        self.out_static = e.out_to_u8(out_nhwc)

    def _capture(self):
        s = torch.cuda.Stream()
        s.wait_stream(torch.cuda.current_stream())
        with torch.cuda.stream(s):
            for _ in range(2):
                self._body()
        torch.cuda.current_stream().wait_stream(s)
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):
                self._body()
            self._graph = g
        except Exception as e:  # noqa: BLE001
            import sys

            print(f"[inferencer] graph capture failed, eager: {e!r}",
                  file=sys.stderr)
            self._use_graph = False

    @torch.no_grad()
    def infer_frame(self, rgb_u8: np.ndarray) -> np.ndarray:
        """uint8 HWC RGB frame -> enhanced uint8 HWC RGB."""
        assert rgb_u8.shape == (self.h, self.w, 3)
        # from_numpy on a read-only array (e.g. PIL-derived) warns once;
        # intentional zero-copy — the tensor is only ever READ by copy_
        self.raw_static.copy_(
            torch.from_numpy(np.ascontiguousarray(rgb_u8)).unsqueeze(0)
        )
        if self._use_graph and self._graph is None:
            self._capture()
        if self._graph is not None:
            self._graph.replay()
        else:
            self._body()
        return self.out_static[0].cpu().numpy()
