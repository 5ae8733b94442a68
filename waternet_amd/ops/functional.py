"""Autograd wrappers for the fused elementwise / layout / reduction kernels.

Reference parity: GatedFusion = net.py:104-108 (K15),
NormalizeNhwc/NormalizeVgg = the TF.normalize calls in train.py:119-123
(K16), mse255_nhwc/Mse255 = the 255-scale MSE reductions in
train.py:125-131 (K18/K19), MaxPool2x2 = the VGG feature maxpools (K17),
NhwcToNchw/NchwToNhwc = layout bridges for the public NCHW contract."""

import torch

from waternet_amd.ops import ext


class NhwcToNchw(torch.autograd.Function):
    """(N,H,W,Cp) bf16 -> (N,C,H,W) fp32 (slices logical channels)."""

    @staticmethod
    def forward(ctx, x, C):
        ctx.Cp = x.size(3)
        return ext().nhwc_to_nchw(x.contiguous(), C)

    @staticmethod
    def backward(ctx, dy):
        return ext().nchw_to_nhwc(dy.contiguous(), ctx.Cp), None


class NchwToNhwc(torch.autograd.Function):
    """(N,C,H,W) fp32 -> (N,H,W,Cp) bf16 zero-padded."""

    @staticmethod
    def forward(ctx, x, Cp):
        ctx.C = x.size(1)
        return ext().nchw_to_nhwc(x.contiguous(), Cp)

    @staticmethod
    def backward(ctx, dy):
        return ext().nhwc_to_nchw(dy.contiguous(), ctx.C), None


class GatedFusion(torch.autograd.Function):
    """out_c = sum_i refined_i_c * map_i (net.py:104-108), NHWC bf16."""

    @staticmethod
    def forward(ctx, maps, rwb, rce, rgc):
        out = ext().fusion_fwd(maps, rwb, rce, rgc)
        ctx.save_for_backward(maps, rwb, rce, rgc)
        return out

    @staticmethod
    def backward(ctx, dout):
        maps, rwb, rce, rgc = ctx.saved_tensors
        dmaps, drwb, drce, drgc = ext().fusion_bwd(dout.contiguous(), maps,
                                                   rwb, rce, rgc)
        return dmaps, drwb, drce, drgc


class MaxPool2x2(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x):
        y, idx = ext().maxpool2x2_fwd(x.contiguous())
        ctx.save_for_backward(idx)
        ctx.hw = (x.size(1), x.size(2))
        return y

    @staticmethod
    def backward(ctx, dy):
        (idx,) = ctx.saved_tensors
        return ext().maxpool2x2_bwd(dy.contiguous(), idx, *ctx.hw)


class NormalizeNhwc(torch.autograd.Function):
    """ImageNet normalize IN the NHWC bf16 layout (full-NHWC loss path:
    WaterNet output -> VGG input with no NCHW round trip)."""

    @staticmethod
    def forward(ctx, x):
        return ext().normalize_nhwc_fwd(x.contiguous())

    @staticmethod
    def backward(ctx, dy):
        return ext().normalize_nhwc_bwd(dy.contiguous())


class NormalizeVgg(torch.autograd.Function):
    """ImageNet normalize fused with NCHW fp32 -> NHWC bf16 (Cp=16)."""

    @staticmethod
    def forward(ctx, x):
        return ext().normalize_vgg_fwd(x.contiguous(), 16)

    @staticmethod
    def backward(ctx, dy):
        return ext().normalize_vgg_bwd(dy.contiguous(), 3)


class Mse255(torch.autograd.Function):
    """mean(square(255*(a-b))) over logical channels; NHWC bf16 inputs."""

    @staticmethod
    def forward(ctx, a, b, clog):
        a = a.contiguous()
        b = b.contiguous()
        s = ext().sqdiff255_sum(a, b, clog)
        numel = (a.numel() // a.size(3)) * clog
        ctx.save_for_backward(a, b)
        ctx.numel = numel
        return (s / numel).to(torch.float32)

    @staticmethod
    def backward(ctx, g):
        a, b = ctx.saved_tensors
        gscale = (g.to(torch.float32) * (2.0 * 255.0 * 255.0 / ctx.numel))
        gscale = gscale.contiguous()
        da = db = None
        if ctx.needs_input_grad[0]:
            da = ext().sqdiff255_bwd(a, b, gscale, 1.0)
        if ctx.needs_input_grad[1]:
            db = ext().sqdiff255_bwd(a, b, gscale, -1.0)
        return da, db, None


def mse255_nhwc(a, b, clog):
    return Mse255.apply(a, b, clog)
