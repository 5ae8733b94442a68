"""Data-parallel training over RCCL/xGMI (or gloo on CPU).

The reference has no distributed code; this subsystem is the
MI355X-native design of SURVEY §5.8: one process per GPU, torch.distributed
with the nccl backend (= RCCL on ROCm) over the node's fully-connected xGMI
links. WaterNet's gradient volume is tiny (~4.36 MB fp32), so the all-reduce
is latency-bound: a SINGLE flat fp32 bucket reduced once per step beats
per-tensor calls by ~38x launch count, and can run asynchronously on RCCL's
stream while metric math proceeds.
"""

import os
from dataclasses import dataclass

import torch
import torch.distributed as dist


@dataclass
class DistEnv:
    rank: int = 0
    local_rank: int = 0
    world_size: int = 1

    @property
    def initialized(self):
        return dist.is_available() and dist.is_initialized()

    def average_metrics(self, metrics: dict) -> dict:
        if not self.initialized or self.world_size == 1:
            return metrics
        keys = sorted(metrics.keys())
        device = (
            torch.device("cuda", self.local_rank)
            if dist.get_backend() == "nccl"
            else torch.device("cpu")
        )
        t = torch.tensor([metrics[k] for k in keys], dtype=torch.float64,
                         device=device)
        dist.all_reduce(t, op=dist.ReduceOp.SUM)
        t /= self.world_size
        return {k: float(v) for k, v in zip(keys, t.tolist())}

    def barrier(self):
        if self.initialized and self.world_size > 1:
            dist.barrier()


def distributed_env() -> DistEnv:
    """Current env without initializing a process group."""
    return DistEnv(
        rank=int(os.environ.get("RANK", 0)),
        local_rank=int(os.environ.get("LOCAL_RANK", 0)),
        world_size=int(os.environ.get("WORLD_SIZE", 1)),
    )


def init_distributed(backend: str = None) -> DistEnv:
    """Initialize torch.distributed from torchrun-style env vars if
    WORLD_SIZE > 1. backend defaults to nccl (RCCL) when CUDA/ROCm devices
    are visible, else gloo."""
    env = distributed_env()
    if env.world_size <= 1 or dist.is_initialized():
        return env
    if backend is None:
        backend = "nccl" if torch.cuda.is_available() else "gloo"
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29500")
    if backend == "nccl":
        torch.cuda.set_device(env.local_rank)
    dist.init_process_group(backend=backend, rank=env.rank,
                            world_size=env.world_size)
    return env


def shard_dataset(dataset, env: DistEnv):
    """Static round-robin shard per rank (keeps per-rank batch count equal
    when len % world_size == 0; drops the ragged tail otherwise)."""
    if env.world_size <= 1:
        return dataset
    n = len(dataset)
    per_rank = n // env.world_size
    idx = list(range(env.rank, per_rank * env.world_size, env.world_size))
    return torch.utils.data.Subset(dataset, idx)


class FlatBucketReducer:
    """Single-flat-bucket gradient all-reduce.

    All model gradients are packed into one contiguous fp32 buffer, reduced
    with ONE collective per step, averaged, and unpacked. With async=True
    the collective is launched on RCCL's stream and waited on just before
    optimizer.step() so it overlaps whatever host/metric work happens in
    between.
    """

    def __init__(self, model: torch.nn.Module, env: DistEnv, async_op=False):
        self.params = [p for p in model.parameters() if p.requires_grad]
        self.env = env
        self.async_op = async_op
        numel = sum(p.numel() for p in self.params)
        device = self.params[0].device if self.params else torch.device("cpu")
        self.flat = torch.zeros(numel, dtype=torch.float32, device=device)
        self._views = []
        offset = 0
        for p in self.params:
            n = p.numel()
            self._views.append(self.flat[offset:offset + n].view(p.shape))
            offset += n
        self._work = None

    def broadcast_params(self):
        if self.env.world_size <= 1:
            return
        offset = 0
        for p, v in zip(self.params, self._views):
            v.copy_(p.data)
            offset += p.numel()
        dist.broadcast(self.flat, src=0)
        for p, v in zip(self.params, self._views):
            p.data.copy_(v)

    def __call__(self):
        """Pack grads -> all-reduce(avg) -> unpack into p.grad."""
        if self.env.world_size <= 1:
            return
        for p, v in zip(self.params, self._views):
            if p.grad is not None:
                v.copy_(p.grad.detach())
            else:
                v.zero_()
        self.flat /= self.env.world_size
        work = dist.all_reduce(self.flat, op=dist.ReduceOp.SUM,
                               async_op=self.async_op)
        if self.async_op:
            self._work = work
        else:
            self._unpack()

    def wait(self):
        if self._work is not None:
            self._work.wait()
            self._work = None
            self._unpack()

    def _unpack(self):
        for p, v in zip(self.params, self._views):
            if p.grad is None:
                p.grad = v.clone()
            else:
                p.grad.detach().copy_(v)
