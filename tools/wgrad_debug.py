"""wgrad debug probes: per-shape parity vs F.conv2d autograd with
wrong-index structure analysis (which k/c/tap regions are bad), a
determinism check, and a big-M variant. (The WN_WGRAD_SPLIT knob it once
drove was removed with the reverted ring-wgrad prototype; the probe
remains useful for any future wgrad work.)"""

import sys
from pathlib import Path

import torch
import torch.nn.functional as F

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from waternet_amd.ops import ext  # noqa: E402
from waternet_amd.ops.conv import pow2_channels  # noqa: E402

SHAPES = [(7, 12, 128), (5, 128, 128), (3, 128, 128), (1, 128, 64),
          (7, 64, 64), (5, 64, 64), (5, 6, 32), (7, 6, 32)]


def ref_dw(dy_nchw, x_nchw, ks):
    w = torch.zeros(dy_nchw.size(1), x_nchw.size(1), ks, ks,
                    device=dy_nchw.device, requires_grad=True)
    y = F.conv2d(x_nchw, w, padding=ks // 2)
    y.backward(dy_nchw)
    return w.grad


def probe(ks, C, K, N, H, W, seed=0):
    e = ext()
    Cp, Kp = pow2_channels(C), pow2_channels(K)
    g = torch.Generator(device="cuda").manual_seed(seed)
    dy = torch.zeros(N, H, W, Kp, device="cuda", dtype=torch.bfloat16)
    x = torch.zeros(N, H, W, Cp, device="cuda", dtype=torch.bfloat16)
    dy[..., :K] = torch.rand(N, H, W, K, generator=g,
                             device="cuda").bfloat16()
    x[..., :C] = torch.rand(N, H, W, C, generator=g, device="cuda").bfloat16()

    dws = []
    for _ in range(2):  # determinism check
        dw = torch.zeros(K, C, ks, ks, device="cuda", dtype=torch.float32)
        e.conv2d_wgrad(dy, x, dw, ks)
        torch.cuda.synchronize()
        dws.append(dw)
    det = torch.equal(dws[0], dws[1])

    ref = ref_dw(dy[..., :K].permute(0, 3, 1, 2).float(),
                 x[..., :C].permute(0, 3, 1, 2).float(), ks)
    dw = dws[0]
    scale = ref.abs().max().item() + 1e-6
    err = (dw - ref).abs() / scale
    bad = err > 0.05
    nbad = int(bad.sum())
    print(f"ks{ks} C{C:3d} K{K:3d} N{N} H{W}: det={det} bad={nbad}"
          f"/{err.numel()} maxerr={err.max().item():.3g}")
    if nbad:
        idx = bad.nonzero()
        ks_bad = sorted(set(idx[:, 0].tolist()))
        cs_bad = sorted(set(idx[:, 1].tolist()))
        taps_bad = sorted(set((idx[:, 2] * ks + idx[:, 3]).tolist()))
        def rng(v):
            return f"[{min(v)}..{max(v)}] n={len(v)}"
        print(f"    bad k {rng(ks_bad)}  c {rng(cs_bad)}  tap {rng(taps_bad)}")
        # rsc structure: rsc = tap*Cp + c -> which BR tile / kblk
        rsc = (idx[:, 2] * ks + idx[:, 3]) * Cp + idx[:, 1]
        kblk = sorted(set((rsc // 16).tolist()))
        print(f"    bad rsc kblk(16) {rng(kblk)}")
        samp = idx[0].tolist()
        print(f"    sample idx {samp}: got {dw[tuple(samp)].item():.4g} "
              f"ref {ref[tuple(samp)].item():.4g}")


def main():
    import os

    print(f"WN_WGRAD_M32={os.environ.get('WN_WGRAD_M32', '(1)')}")
    for ks, C, K in SHAPES:
        probe(ks, C, K, 2, 16, 16)
    print("--- big M ---")
    for ks, C, K in [(5, 128, 128), (1, 128, 64)]:
        probe(ks, C, K, 16, 112, 112)


if __name__ == "__main__":
    main()
