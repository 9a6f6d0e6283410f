"""rocTX range markers for rocprofv3 timelines.

`torch.cuda.nvtx` maps to rocTX on ROCm, so `rocprofv3 --marker-trace`
shows these ranges alongside the kernel trace (the reference
had no tracing at all).  No-ops on CPU.
"""
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


def mark(name: str):
    if _ENABLED:
        torch.cuda.nvtx.mark(name)
