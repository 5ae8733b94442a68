"""Native CDNA4 forward paths for WaterNet and the VGG perceptual model.

The nn.Module graphs keep their reference-exact parameter schema; this module
runs their forward passes through the hand-written HIP kernels: fused input
builders (cat folding), NHWC bf16 MFMA implicit-GEMM convolutions with fused
bias+activation, the gated-fusion kernel, and 2x2 maxpool. Autograd flows
through waternet_amd.ops Functions, so `.backward()` uses the HIP
dgrad/wgrad kernels.
"""

import torch
from torch import nn

from waternet_amd.ops import ext
from waternet_amd.ops.conv import ACT_RELU, ACT_SIGMOID, ConvSpec, conv_bias_act
from waternet_amd.ops.functional import (
    GatedFusion,
    MaxPool2x2,
    NchwToNhwc,
    NhwcToNchw,
)


class WaterNetNativeState:
    """Per-model ConvSpec table (packed-weight caches)."""

    def __init__(self, model):
        cmg = model.cmg
        acts = [ACT_RELU] * 7 + [ACT_SIGMOID]
        self.cmg_specs = [
            ConvSpec(getattr(cmg, f"conv{i + 1}"), acts[i]) for i in range(8)
        ]
        self.refiner_specs = {}
        for name in ("wb_refiner", "ce_refiner", "gc_refiner"):
            ref = getattr(model, name)
            self.refiner_specs[name] = [
                ConvSpec(getattr(ref, f"conv{i + 1}"), ACT_RELU)
                for i in range(3)
            ]

    def mark_dirty(self):
        for s in self.cmg_specs:
            s.mark_dirty()
        for specs in self.refiner_specs.values():
            for s in specs:
                s.mark_dirty()


def _state(model) -> WaterNetNativeState:
    st = getattr(model, "_wn_native_state", None)
    if st is None:
        st = WaterNetNativeState(model)
        model._wn_native_state = st
    return st


def waternet_forward_from_inputs(model, cmg_in, rwb_in, rce_in, rgc_in):
    """Core NHWC pipeline from pre-built (cat-folded) conv inputs:
    8 CMG convs -> 3x3 refiner convs -> gated fusion (net.py:104-108).
    Inputs/output (N,H,W,16) bf16, 3 logical output channels — the
    full-NHWC training/inference paths stay in this layout end to end."""
    st = _state(model)
    t = cmg_in
    for spec in st.cmg_specs:
        t = conv_bias_act(t, spec)
    maps = t

    refined = []
    for name, rin in (("wb_refiner", rwb_in), ("ce_refiner", rce_in),
                      ("gc_refiner", rgc_in)):
        r = rin
        for spec in st.refiner_specs[name]:
            r = conv_bias_act(r, spec)
        refined.append(r)

    return GatedFusion.apply(maps, refined[0], refined[1], refined[2])


def waternet_forward_native(model, x, wb, ce, gc):
    """x, wb, ce, gc: (N,3,H,W) fp32 CUDA -> (N,3,H,W) fp32.

    Kernel pipeline: build_inputs (fused cat, net.py:46/76) ->
    waternet_forward_from_inputs -> NHWC->NCHW (public NCHW contract)."""
    xc = x.float().contiguous()
    inputs = ext().build_inputs(xc, wb.float().contiguous(),
                                ce.float().contiguous(),
                                gc.float().contiguous())
    out_nhwc = waternet_forward_from_inputs(model, *inputs)
    return NhwcToNchw.apply(out_nhwc, 3)


class VggNativeState:
    def __init__(self, vgg_model):
        self.plan = []  # list of ("conv", ConvSpec) / ("pool", None)
        for m in vgg_model.model:
            if isinstance(m, nn.Conv2d):
                self.plan.append(("conv", ConvSpec(m, ACT_RELU)))
            elif isinstance(m, nn.MaxPool2d):
                self.plan.append(("pool", None))
            elif isinstance(m, nn.ReLU):
                pass  # fused into the conv epilogue
            else:
                raise RuntimeError(f"unexpected VGG layer {type(m)}")

    def mark_dirty(self):
        for kind, spec in self.plan:
            if kind == "conv":
                spec.mark_dirty()


def vgg_state(vgg_model) -> VggNativeState:
    st = getattr(vgg_model, "_wn_native_state", None)
    if st is None:
        st = VggNativeState(vgg_model)
        vgg_model._wn_native_state = st
    return st


def vgg_forward_nhwc(vgg_model, x_nhwc):
    """(N,H,W,16) bf16 (already ImageNet-normalized) ->
    (N,H/16,W/16,512) bf16 — the full-NHWC perceptual tower."""
    t = x_nhwc
    for kind, spec in vgg_state(vgg_model).plan:
        if kind == "conv":
            t = conv_bias_act(t, spec)
        else:
            t = MaxPool2x2.apply(t)
    return t


def vgg_prepack(vgg_model):
    """Force-build the ConvSpec table and pack all 16 VGG conv weights ON
    THE CURRENT STREAM. The frozen VGG never repacks afterwards, so a step
    engine that runs the fy tower on a side stream must call this once from
    the main stream (otherwise the first step's fx tower reads buffers
    packed on the side stream without ordering)."""
    for kind, spec in vgg_state(vgg_model).plan:
        if kind == "conv":
            spec.refresh_if_needed()


def vgg_forward_native(vgg_model, x):
    """x: (N,3,H,W) fp32 (already ImageNet-normalized by the caller) ->
    (N,512,H/16,W/16) fp32."""
    t = NchwToNhwc.apply(x, 16)
    t = vgg_forward_nhwc(vgg_model, t)
    return NhwcToNchw.apply(t, 512)
