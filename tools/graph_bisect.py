"""Find which stage of the train step breaks hipGraph capture (run on GPU)."""
import sys
import torch

sys.path.insert(0, ".")
from waternet_amd.engine.fast import BenchTrainer  # noqa: E402
from waternet_amd.models.vgg import normalize_imagenet  # noqa: E402
from waternet_amd.ops.preprocess import gpu_transform_batch  # noqa: E402
from waternet_amd.ops import ext  # noqa: E402
from waternet_amd.ops.ssim import ssim_native  # noqa: E402

tr = BenchTrainer(batch_size=4, height=64, width=64, device="cuda:0",
                  use_graph=False, seed=0)
tr.step()  # warmup everything incl. one full eager step
torch.cuda.synchronize()

e = ext()


def stage_preprocess():
    wb, gc, he = gpu_transform_batch(tr.raw_static)
    return wb, gc, he


def stage_convert():
    wb, gc, he = stage_preprocess()
    return (e.u8_to_nchw(tr.raw_static), e.u8_to_nchw(wb), e.u8_to_nchw(gc),
            e.u8_to_nchw(he), e.u8_to_nchw(tr.ref_static))


def stage_forward():
    raw_f, wb_f, gc_f, he_f, ref_f = stage_convert()
    return tr.model(raw_f, wb_f, he_f, gc_f), ref_f


def stage_loss():
    out, ref_f = stage_forward()
    fx = tr.vgg(normalize_imagenet(out))
    with torch.no_grad():
        fy = tr.vgg(normalize_imagenet(ref_f))
    dp = 255.0 * (fx - fy)
    p = torch.mean(dp * dp)
    dm = 255.0 * (out - ref_f)
    return 0.05 * p + torch.mean(dm * dm), out, ref_f, p

def stage_backward():
    loss, out, ref_f, p = stage_loss()
    tr.opt.zero_grad()
    loss.backward()
    return loss, out, ref_f, p

def stage_adam():
    r = stage_backward()
    tr.opt.step()
    return r

def stage_metrics():
    loss, out, ref_f, p = stage_adam()
    with torch.no_grad():
        ssim = ssim_native(out.detach(), ref_f, 1.0)
        mse01 = torch.mean((out.detach() - ref_f) ** 2)
        psnr = 10.0 * torch.log10(1.0 / mse01)
        tr.metric_sums += torch.stack([
            loss.detach().double(), p.detach().double(),
            loss.detach().double(), ssim.double(), psnr.double()])


stages = [("preprocess", stage_preprocess), ("convert", stage_convert),
          ("forward", stage_forward), ("loss", stage_loss),
          ("backward", stage_backward), ("adam", stage_adam),
          ("metrics", stage_metrics)]

for name, fn in stages:
    # warmup on side stream
    s = torch.cuda.Stream()
    s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        fn()
    torch.cuda.current_stream().wait_stream(s)
    torch.cuda.synchronize()
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        g.replay()
        torch.cuda.synchronize()
        print(f"CAPTURE OK: {name}")
    except Exception as ex:  # noqa: BLE001
        print(f"CAPTURE FAIL at {name}: {ex!r}"[:400])
        break
