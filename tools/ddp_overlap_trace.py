"""Produce trace evidence that the DDP flat-arena all-reduce runs on the
comm stream OVERLAPPED with the SSIM/PSNR metric kernels (SURVEY §5.8
design). Run under rocprofv3 --kernel-trace with a world-1 RCCL group and
the engine's world>1 branch forced (collective = identity)."""

import os
import sys
from pathlib import Path

import numpy as np
import torch
import torch.distributed as dist

sys.path.insert(0, str(Path(__file__).resolve().parent.parent))

from waternet_amd.engine.fast import FastStepEngine  # noqa: E402
from waternet_amd.models.waternet import WaterNet  # noqa: E402

os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
os.environ.setdefault("MASTER_PORT", "29881")
torch.cuda.set_device(0)
dist.init_process_group("nccl", rank=0, world_size=1)

torch.manual_seed(0)
model = WaterNet().to("cuda:0")
eng = FastStepEngine(model, batch_size=16, height=112, width=112,
                     device="cuda:0", use_graph=False, world_size=1)
dist.broadcast(eng.opt.master, src=0)
eng.world = 2  # exercise the comm-stream all-reduce branch (identity)
rng = np.random.default_rng(0)
raw = torch.from_numpy(rng.integers(0, 256, (16, 112, 112, 3),
                                    dtype=np.uint8)).to("cuda:0")
ref = torch.from_numpy(rng.integers(0, 256, (16, 112, 112, 3),
                                    dtype=np.uint8)).to("cuda:0")
for _ in range(6):
    eng.load_batch(raw, ref)
    eng.step()
torch.cuda.synchronize()
print("steps done:", eng._steps, eng.metrics())
dist.destroy_process_group()
