"""Conv2d (stride-1, same-pad) over the CDNA4 MFMA implicit-GEMM kernels —
every nn.Conv2d the reference runs (net.py:14-45,66-71 and the VGG19
features) dispatches here on GPU.

ConvBiasAct is the autograd wrapper around csrc/conv_mfma.hip: forward runs
the fused conv+bias+activation kernel on NHWC bf16; backward runs
activation-backward, dgrad (the same implicit-GEMM kernel on rotated/
transposed packed weights) and wgrad/bias-grad (atomic fp32 accumulation).

Weight packing (NCHW fp32 master -> bf16 [Kp][RS*Cp] fwd / [Cp][RS*Kp]
dgrad layouts) is cached per layer in ConvSpec and refreshed when the
parameter version changes (or when FusedAdam marks it dirty).

Gradient routing: if the parameter carries a `_wn_grad_view` attribute
(attached by waternet_amd.ops.adam.FusedAdam's flat gradient arena), wgrad
accumulates into that pre-zeroed view in place and the Function returns None
(the arena already holds the sum); otherwise a fresh zero tensor is
allocated and returned for standard autograd accumulation.
"""

import torch
from torch import nn

from waternet_amd.ops import ext

ACT_NONE, ACT_RELU, ACT_SIGMOID = 0, 1, 2


def pow2_channels(c: int) -> int:
    p = 16
    while p < c:
        p *= 2
    return p


class ConvSpec:
    """Per-layer metadata + packed-weight cache for one nn.Conv2d."""

    def __init__(self, module: nn.Conv2d, act: int):
        K, C, R, S = module.weight.shape
        assert R == S, "square kernels only"
        self.mod = module
        self.act = act
        self.ks = R
        self.K, self.C = K, C
        self.Kp = pow2_channels(K)
        self.Cp = pow2_channels(C)
        self._wp = None
        self._wd = None
        self._version = None

    def refresh_if_needed(self):
        w = self.mod.weight
        ver = (w._version, w.data_ptr())
        if self._wp is None or ver != self._version:
            wc = w.data.contiguous()
            if self._wp is None:
                self._wp = ext().pack_weight_fwd(wc, self.Kp, self.Cp)
                self._wd = ext().pack_weight_dgrad(wc, self.Kp, self.Cp)
            else:
                # Rewrite IN PLACE: pack_all descriptors bake these device
                # pointers, so the buffers must stay at the same address
                # for the life of the spec (never reallocate).
                self._wp.copy_(ext().pack_weight_fwd(wc, self.Kp, self.Cp))
                self._wd.copy_(ext().pack_weight_dgrad(wc, self.Kp, self.Cp))
            self._version = ver

    def mark_dirty(self):
        self._version = None

    def packed_fwd(self):
        self.refresh_if_needed()
        return self._wp

    def packed_dgrad(self):
        self.refresh_if_needed()
        return self._wd


class ConvBiasAct(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, weight, bias, spec: ConvSpec):
        # x: (N, H, W, Cp) bf16 contiguous
        y = ext().conv2d_fwd(x, spec.packed_fwd(), bias, spec.ks, spec.Kp,
                             spec.K, spec.act)
        ctx.save_for_backward(x, y)
        ctx.spec = spec
        return y

    @staticmethod
    def backward(ctx, dy):
        x, y = ctx.saved_tensors
        spec = ctx.spec
        dy = dy.contiguous()
        need_x, need_w, need_b = ctx.needs_input_grad[:3]
        db_done = False
        bias_view = (getattr(spec.mod.bias, "_wn_grad_view", None)
                     if spec.mod.bias is not None else None)
        if (spec.act != ACT_NONE and need_b and bias_view is not None):
            # fused: dpre = dy*act'(y) with the bias column sums
            # accumulated in the same pass (saves a full dpre re-read)
            dpre = ext().act_bwd_bias(dy, y, spec.act, bias_view)
            db_done = True
        elif spec.act != ACT_NONE:
            dpre = ext().act_bwd(dy, y, spec.act)
        else:
            dpre = dy

        dx = dw_ret = db_ret = None
        if need_x:
            # dgrad: conv of dpre (channels Kp) with rotated/transposed
            # weights; output logical channels = C.
            dx = ext().conv2d_fwd(dpre, spec.packed_dgrad(), None, spec.ks,
                                  spec.Cp, spec.C, ACT_NONE)
        if need_w:
            view = getattr(spec.mod.weight, "_wn_grad_view", None)
            if view is not None:
                ext().conv2d_wgrad(dpre, x, view, spec.ks)
            else:
                dw_ret = torch.zeros_like(spec.mod.weight,
                                          memory_format=torch.contiguous_format)
                ext().conv2d_wgrad(dpre, x, dw_ret, spec.ks)
        if need_b and spec.mod.bias is not None and not db_done:
            view = getattr(spec.mod.bias, "_wn_grad_view", None)
            if view is not None:
                ext().bias_grad(dpre, view)
            else:
                db_ret = torch.zeros_like(spec.mod.bias)
                ext().bias_grad(dpre, db_ret)
        return dx, dw_ret, db_ret, None


def conv_bias_act(x, spec: ConvSpec):
    return ConvBiasAct.apply(x, spec.mod.weight, spec.mod.bias, spec)


def build_pack_descriptor(specs, device):
    """int64 [n][9] descriptor for ext().pack_all over a list of ConvSpecs.
    Forces allocation of each spec's packed buffers first (pointers must be
    stable — they are: buffers are allocated once and rewritten in place)."""
    import torch as _torch

    rows = []
    for s in specs:
        s.refresh_if_needed()
        w = s.mod.weight
        rows.append([
            w.data.data_ptr(), s._wp.data_ptr(), s._wd.data_ptr(),
            s.K, s.C, s.ks, s.ks, s.Kp, s.Cp,
        ])
    return _torch.tensor(rows, dtype=_torch.int64, device=device)


def mark_specs_packed(specs):
    """Record current weight versions on every spec so the lazy per-layer
    refresh skips (the batched pack_all just rewrote the buffers)."""
    for s in specs:
        w = s.mod.weight
        s._version = (w._version, w.data_ptr())
