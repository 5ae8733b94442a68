"""Multi-process (gloo, world_size=2) tests of the distributed layer on CPU.

The RCCL path on MI355X uses the same torch.distributed code with the nccl
backend; these tests exercise the collective logic itself.
"""

import os

import pytest
import torch
import torch.distributed as dist
import torch.multiprocessing as mp


def _run_allreduce_worker(rank, world_size, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["LOCAL_RANK"] = str(rank)
    try:
        from waternet_amd.models.waternet import WaterNet
        from waternet_amd.parallel import FlatBucketReducer, init_distributed

        env = init_distributed(backend="gloo")
        torch.manual_seed(rank)  # different init per rank on purpose
        model = WaterNet()
        reducer = FlatBucketReducer(model, env)
        reducer.broadcast_params()

        # After broadcast all ranks hold rank0's params
        psum = sum(p.double().sum().item() for p in model.parameters())

        # Different per-rank input -> different grads; all-reduce averages
        torch.manual_seed(100 + rank)
        x = torch.rand(2, 3, 32, 32)
        out = model(x, x, x, x)
        out.mean().backward()
        reducer()

        g0 = model.cmg.conv1.weight.grad.clone()
        # gather grads to rank 0 and verify all equal
        gathered = [torch.zeros_like(g0) for _ in range(world_size)]
        dist.all_gather(gathered, g0)
        grads_equal = all(
            torch.allclose(gathered[0], g, atol=1e-7) for g in gathered
        )

        metrics = env.average_metrics({"m": float(rank)})
        q.put((rank, psum, grads_equal, metrics["m"]))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, "ERROR", repr(e), None))


import pytest


@pytest.mark.parametrize("world_size,port", [(2, 29811), (4, 29815)])
def test_flat_bucket_reducer(world_size, port):
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_run_allreduce_worker, args=(r, world_size, port, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world_size)]
    for p in procs:
        p.join(timeout=120)
    errors = [r for r in results if r[1] == "ERROR"]
    assert not errors, errors
    psums = {r[1] for r in results}
    assert len({round(s, 6) for s in psums}) == 1, "broadcast_params failed"
    assert all(r[2] for r in results), "grads differ across ranks"
    expect = sum(range(world_size)) / world_size
    assert all(abs(r[3] - expect) < 1e-9 for r in results), "metric avg wrong"


def _run_shard_worker(rank, world_size, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        from waternet_amd.parallel import init_distributed, shard_dataset

        env = init_distributed(backend="gloo")
        ds = list(range(10))
        shard = shard_dataset(ds, env)
        q.put((rank, list(shard)))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, repr(e)))


def test_shard_dataset_world2():
    world_size = 2
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_run_shard_worker, args=(r, world_size, 29812, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = dict(q.get() for _ in range(world_size))
    for p in procs:
        p.join(timeout=60)
    assert results[0] == [0, 2, 4, 6, 8]
    assert results[1] == [1, 3, 5, 7, 9]


def _run_parity_worker(rank, world_size, port, q):
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        from waternet_amd.models.waternet import WaterNet
        from waternet_amd.parallel import FlatBucketReducer, init_distributed

        env = init_distributed(backend="gloo")
        torch.manual_seed(0)
        model = WaterNet()
        opt = torch.optim.Adam(model.parameters(), lr=1e-3)
        reducer = FlatBucketReducer(model, env)
        reducer.broadcast_params()

        torch.manual_seed(42)  # same data on every rank, then shard
        x = torch.rand(2 * world_size, 3, 16, 16)
        ref = torch.rand(2 * world_size, 3, 16, 16)
        xs = x[rank * 2:(rank + 1) * 2]
        refs = ref[rank * 2:(rank + 1) * 2]
        for _ in range(2):
            out = model(xs, xs, xs, xs)
            loss = torch.mean((out - refs) ** 2)
            opt.zero_grad()
            loss.backward()
            reducer()
            opt.step()
        w = model.cmg.conv1.weight.detach().clone()
        q.put((rank, w.numpy().tolist()))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, repr(e)))


def test_ddp_loss_parity_vs_single():
    """2-rank DDP with the flat-bucket reducer must match a single process
    training on the full batch (same seeds, same data) — SURVEY §4's
    distributed parity requirement."""
    world_size = 2
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_run_parity_worker, args=(r, world_size, 29813, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = dict(q.get() for _ in range(world_size))
    for p in procs:
        p.join(timeout=120)
    results = {r: torch.tensor(w) for r, w in results.items()}

    # single-process reference on the full batch
    from waternet_amd.models.waternet import WaterNet

    torch.manual_seed(0)
    model = WaterNet()
    opt = torch.optim.Adam(model.parameters(), lr=1e-3)
    torch.manual_seed(42)
    x = torch.rand(4, 3, 16, 16)
    ref = torch.rand(4, 3, 16, 16)
    for _ in range(2):
        out = model(x, x, x, x)
        loss = torch.mean((out - ref) ** 2)
        opt.zero_grad()
        loss.backward()
        opt.step()
    w_single = model.cmg.conv1.weight.detach()
    for r, w in results.items():
        assert torch.allclose(w, w_single, atol=1e-5), (
            r, (w - w_single).abs().max())


def _run_arena_ddp_worker(rank, world_size, port, q):
    """bench.py's exact DDP arithmetic on CPU/gloo: FusedAdam flat arena,
    rank-0 master broadcast, grads.div_(world) + SUM all-reduce before
    opt.step() — the path the driver's 8-GPU scale run exercises."""
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    os.environ["RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    try:
        from waternet_amd.models.waternet import WaterNet
        from waternet_amd.ops.adam import FusedAdam
        from waternet_amd.parallel import init_distributed

        init_distributed(backend="gloo")
        torch.manual_seed(1234 + rank)  # per-rank init, as bench.py
        model = WaterNet()
        opt = FusedAdam(model.parameters(), lr=1e-3, model=model)
        dist.broadcast(opt.master, src=0)  # bench.py:128

        torch.manual_seed(500 + rank)  # per-rank data shard
        x = torch.rand(2, 3, 16, 16)
        ref = torch.rand(2, 3, 16, 16)
        for _ in range(2):
            out = model(x, x, x, x)
            loss = torch.mean((255.0 * (out - ref)) ** 2)
            opt.zero_grad()
            loss.backward()
            opt.grads.div_(world_size)  # fast.py DDP branch
            dist.all_reduce(opt.grads)
            opt.step()
        q.put((rank, opt.master.double().sum().item(),
               model.cmg.conv1.weight.detach().numpy().tolist()))
        dist.destroy_process_group()
    except Exception as e:  # noqa: BLE001
        q.put((rank, "ERROR", repr(e)))


@pytest.mark.parametrize("world_size,port", [(2, 29817), (4, 29818),
                                             (8, 29819)])
def test_bench_arena_ddp_flow(world_size, port):
    """The driver's scale bench runs this exact arithmetic at
    --nproc-per-node 8; world 8 on CPU/gloo proves the 8-rank flow
    (per-rank init -> rank-0 master broadcast -> grads/8 + SUM all-reduce
    -> step keeps all ranks bit-identical) before any 8-GPU hardware."""
    ctx = mp.get_context("spawn")
    q = ctx.SimpleQueue()
    procs = [
        ctx.Process(target=_run_arena_ddp_worker,
                    args=(r, world_size, port, q))
        for r in range(world_size)
    ]
    for p in procs:
        p.start()
    results = [q.get() for _ in range(world_size)]
    for p in procs:
        p.join(timeout=120)
    errors = [r for r in results if r[1] == "ERROR"]
    assert not errors, errors
    # after broadcast + identical averaged grads, ranks stay bit-identical
    sums = {round(r[1], 9) for r in results}
    assert len(sums) == 1, results
    w = {r[0]: torch.tensor(r[2]) for r in results}
    for r in range(1, world_size):
        assert torch.equal(w[0], w[r]), f"rank {r} diverged"
