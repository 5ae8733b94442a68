"""RCCL-on-hardware tests: execute the nccl(=RCCL)
backend on the MI355X — world-size-1 process group init, a real collective
on a CUDA tensor, and the fast engine's DDP step path (flat-arena
all-reduce on the comm stream) — so the first 8-GPU scale run is not the
first time this code touches RCCL.
"""

import os

import numpy as np
import pytest
import torch
import torch.distributed as dist

pytestmark = pytest.mark.gpu

DEV = "cuda:0"


@pytest.fixture()
def nccl_world1():
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29871")
    torch.cuda.set_device(0)
    dist.init_process_group("nccl", rank=0, world_size=1)
    yield
    dist.destroy_process_group()


def test_nccl_world1_collectives(nccl_world1):
    t = torch.full((1024,), 3.0, device=DEV)
    dist.all_reduce(t)  # world 1: identity, but a REAL RCCL kernel
    torch.cuda.synchronize()
    assert torch.all(t == 3.0)
    dist.broadcast(t, src=0)
    dist.barrier()
    g = [torch.zeros_like(t)]
    dist.all_gather(g, t)
    assert torch.all(g[0] == 3.0)


def test_nccl_fast_engine_ddp_step(nccl_world1):
    """The DDP branch of the fast step (grads.div_ + all_reduce on the comm
    stream, overlapped with metrics) executed against real RCCL. World
    size 1 makes the collective an identity, so the numbers must match a
    world-1 engine exactly apart from the grads/world scaling."""
    from waternet_amd.engine.fast import FastStepEngine
    from waternet_amd.models.waternet import WaterNet

    torch.manual_seed(11)
    model = WaterNet().to(DEV)
    eng = FastStepEngine(model, batch_size=2, height=64, width=64,
                         device=DEV, use_graph=False, world_size=1)
    dist.broadcast(eng.opt.master, src=0)  # the bench.py DDP init path

    rng = np.random.default_rng(0)
    raw = torch.from_numpy(
        rng.integers(0, 256, size=(2, 64, 64, 3), dtype=np.uint8))
    ref = torch.from_numpy(
        rng.integers(0, 256, size=(2, 64, 64, 3), dtype=np.uint8))
    eng.load_batch(raw.to(DEV), ref.to(DEV))
    eng.step()  # baseline world-1 step

    # Force the world>1 code path: with world 2 the grads are halved then
    # all-reduced (identity at world 1) -> a finite, RCCL-executing step.
    eng.world = 2
    eng.load_batch(raw.to(DEV), ref.to(DEV))
    eng.step()
    torch.cuda.synchronize()
    m = eng.metrics()
    assert all(np.isfinite(v) for v in m.values()), m
