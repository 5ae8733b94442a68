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


def _engine_ddp_worker(rank, world_size, port, q):
    """Full FastStepEngine DDP step at world_size=2 — BOTH ranks on the one
    GPU (gloo backend moves the CUDA grad arena through host staging; the
    engine code path — comm-stream all_reduce overlapped with metrics,
    grads.div_(world), master broadcast — is exactly bench.py's)."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        torch.cuda.set_device(0)
        dist.init_process_group("gloo", rank=rank, world_size=world_size)
        from waternet_amd.engine.fast import FastStepEngine
        from waternet_amd.models.waternet import WaterNet

        torch.manual_seed(1234 + rank)  # per-rank init, as bench.py
        model = WaterNet().to(DEV)
        eng = FastStepEngine(model, batch_size=2, height=64, width=64,
                             device=DEV, use_graph=False,
                             world_size=world_size)
        dist.broadcast(eng.opt.master, src=0)

        rng = np.random.default_rng(500 + rank)  # per-rank data shard
        raw = torch.from_numpy(rng.integers(
            0, 256, size=(2, 64, 64, 3), dtype=np.uint8)).to(DEV)
        ref = torch.from_numpy(rng.integers(
            0, 256, size=(2, 64, 64, 3), dtype=np.uint8)).to(DEV)
        for _ in range(2):
            eng.load_batch(raw, ref)
            eng.step()
        torch.cuda.synchronize()
        m = eng.metrics()
        q.put((rank, round(eng.opt.master.double().sum().item(), 6),
               all(np.isfinite(v) for v in m.values())))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, "ERROR", repr(e)))


def test_engine_ddp_world2_one_gpu():
    """DDP invariant on hardware: after master broadcast + averaged-grad
    all-reduce, both ranks' flat master arenas stay identical across
    steps."""
    import torch.multiprocessing as mp

    world_size = 2
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [ctx.Process(target=_engine_ddp_worker,
                         args=(r, world_size, 29873, q))
             for r in range(world_size)]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world_size)]
    for p in procs:
        p.join(timeout=300)
    errors = [r for r in results if r[1] == "ERROR"]
    assert not errors, errors
    assert all(r[2] for r in results), f"non-finite metrics: {results}"
    masters = {r[1] for r in results}
    assert len(masters) == 1, f"rank masters diverged: {results}"
