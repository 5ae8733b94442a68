"""roctx range helpers: annotate dataload/preprocess/forward/
loss/backward/allreduce/optimizer phases so rocprofv3 --sys-trace attributes
kernel time to pipeline stages. torch.cuda.nvtx maps to roctx on ROCm;
no-ops cleanly on CPU."""

from contextlib import contextmanager

import torch


_ENABLED = torch.cuda.is_available()


@contextmanager
def trace_range(name: str):
    if _ENABLED:
        torch.cuda.nvtx.range_push(name)
        try:
            yield
        finally:
            torch.cuda.nvtx.range_pop()
    else:
        yield
