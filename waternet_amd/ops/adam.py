"""Fused flat-buffer Adam for MI355X.

All parameters live in ONE contiguous fp32 master buffer (p.data re-pointed
to views), gradients accumulate into ONE flat fp32 arena (the conv wgrad /
bias-grad kernels atomically add straight into the arena views attached as
p._wn_grad_view), and the update is a single k_adam kernel over the flat
buffers — 3 kernels per step (arena zero + backward writes + adam) instead
of ~38 per-tensor optimizer launches.

The flat arena is also the DDP gradient bucket: FlatBucketReducer-style
all-reduce of `self.grads` is one latency-bound RCCL collective per step
(SURVEY §5.8).

Matches torch.optim.Adam numerics (bias-corrected, eps outside sqrt-hat).
Exposes the param_groups / state_dict surface that StepLR needs.
"""

import torch

from waternet_amd.ops import ext, native_available


class FusedAdam(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 model=None):
        params = list(params)
        defaults = dict(lr=lr, betas=betas, eps=eps)
        super().__init__(params, defaults)
        plist = [p for g in self.param_groups for p in g["params"]]
        assert all(p.dtype == torch.float32 for p in plist), \
            "FusedAdam: fp32 master params only"
        total = sum(p.numel() for p in plist)
        dev = plist[0].device
        self.master = torch.empty(total, dtype=torch.float32, device=dev)
        self.grads = torch.zeros(total, dtype=torch.float32, device=dev)
        self.exp_avg = torch.zeros(total, dtype=torch.float32, device=dev)
        self.exp_avg_sq = torch.zeros(total, dtype=torch.float32, device=dev)
        self._plist = plist
        self._views = []
        off = 0
        for p in plist:
            n = p.numel()
            mv = self.master[off:off + n].view_as(p)
            mv.copy_(p.data)
            p.data = mv  # param IS a view of the master buffer
            gv = self.grads[off:off + n].view_as(p)
            p._wn_grad_view = gv
            self._views.append(gv)
            off += n
        self._step = 0
        self._model = model
        self._use_native = dev.type == "cuda" and native_available()
        # device-resident hyperparameters: a hipGraph-captured step reads
        # these buffers, so bias correction and StepLR changes stay correct
        # under graph replay.
        self.lr_buf = torch.tensor([lr], dtype=torch.float32, device=dev)
        self.step_buf = torch.zeros(1, dtype=torch.int32, device=dev)
        self._lr_synced = lr

    def zero_grad(self, set_to_none: bool = True):  # noqa: ARG002
        self.grads.zero_()
        for p, gv in zip(self._plist, self._views):
            # Pre-set p.grad to the arena view: backward kernels accumulate
            # into it in place and return None, so autograd leaves it as-is.
            p.grad = gv

    def sync_lr(self):
        """Copy param_groups lr into the device buffer if it changed (call
        outside graph replay; StepLR mutates param_groups on the host)."""
        lr = self.param_groups[0]["lr"]
        if lr != self._lr_synced:
            self.lr_buf.fill_(lr)
            self._lr_synced = lr

    @torch.no_grad()
    def step(self, closure=None):  # noqa: ARG002
        self._step += 1
        group = self.param_groups[0]
        b1, b2 = group["betas"]
        if self._use_native:
            self.sync_lr()
            ext().adam_step(self.master, self.grads, self.exp_avg,
                            self.exp_avg_sq, self.lr_buf, b1, b2,
                            group["eps"], self.step_buf)
        else:
            g = self.grads
            self.exp_avg.mul_(b1).add_(g, alpha=1 - b1)
            self.exp_avg_sq.mul_(b2).addcmul_(g, g, value=1 - b2)
            bc1 = 1 - b1**self._step
            bc2 = 1 - b2**self._step
            denom = (self.exp_avg_sq / bc2).sqrt_().add_(group["eps"])
            self.master.addcdiv_(self.exp_avg / bc1, denom, value=-group["lr"])
        self._mark_model_dirty()

    def _mark_model_dirty(self):
        # Master changed underneath the params (no tensor version bump):
        # invalidate packed-weight caches.
        if self._model is not None:
            st = getattr(self._model, "_wn_native_state", None)
            if st is not None:
                st.mark_dirty()

    def state_dict(self):
        sd = super().state_dict()
        sd["wn_fused"] = {
            "step": self._step,
            "exp_avg": self.exp_avg,
            "exp_avg_sq": self.exp_avg_sq,
        }
        return sd

    def load_state_dict(self, sd):
        fused = sd.pop("wn_fused", None)
        super().load_state_dict(sd)
        if fused is not None:
            self._step = fused["step"]
            self.step_buf.fill_(self._step)
            self.exp_avg.copy_(fused["exp_avg"])
            self.exp_avg_sq.copy_(fused["exp_avg_sq"])
