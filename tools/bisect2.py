import sys, traceback
import torch
sys.path.insert(0, ".")
from waternet_amd.engine.fast import BenchTrainer
from waternet_amd.models.vgg import normalize_imagenet
tr = BenchTrainer(batch_size=4, height=64, width=64, device="cuda:0", use_graph=False, seed=0)
tr.step(); torch.cuda.synchronize()

def cap(name, fn):
    s = torch.cuda.Stream(); s.wait_stream(torch.cuda.current_stream())
    with torch.cuda.stream(s):
        fn()
    torch.cuda.current_stream().wait_stream(s)
    try:
        g = torch.cuda.CUDAGraph()
        with torch.cuda.graph(g):
            fn()
        print("OK:", name)
        return True
    except Exception:
        print("FAIL:", name)
        traceback.print_exc(limit=8)
        return False

from waternet_amd.ops import ext
e = ext()
raw_f = e.u8_to_nchw(tr.raw_static)
out = tr.model(raw_f, raw_f, raw_f, raw_f).detach()
cap("normalize", lambda: normalize_imagenet(out))
nrm = normalize_imagenet(out).detach()
cap("vgg_nograd", lambda: tr.vgg(nrm))
x = nrm.requires_grad_(False)
cap("vgg_grad_input", lambda: tr.vgg(nrm.clone().requires_grad_(True)))
