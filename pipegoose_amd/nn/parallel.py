"""Base class for parallelization wrappers.

Reference parity: pipegoose/nn/parallel.py:32-93 — after ``parallelize()`` the
model carries metadata mapping its rank tuple to a local device, and ``.to()``/
``.cuda()`` are patched so ``model.to("cuda")`` lands parameters on the right
local HIP device (one process per GPU).
"""
from abc import ABC, abstractmethod
from dataclasses import dataclass
from functools import partial

import torch
from torch import nn

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


@dataclass
class ParallelMetadata:
    device: int
    local_device: int


class Parallel(ABC):
    def __init__(self, module: nn.Module, parallel_context: ParallelContext):
        self.module = module
        self.parallel_context = parallel_context

    @abstractmethod
    def parallelize(self) -> nn.Module:
        ...

    def deparallelize(self) -> nn.Module:
        raise NotImplementedError

    def _save_metadata(self, module: nn.Module, parallel_context: ParallelContext):
        local_rank = parallel_context.get_local_rank(ParallelMode.GLOBAL) % max(
            torch.cuda.device_count(), 1
        ) if torch.cuda.is_available() else 0

        metadata = ParallelMetadata(device=local_rank, local_device=local_rank)
        setattr(module, "parallel_metadata", metadata)
        module.to = partial(_to_device, module)
        module.cuda = partial(_to_cuda, module)


def _to_device(module: nn.Module, device: str = None, *args, **kwargs):
    """Patched ``.to()``: route any 'cuda' request to this rank's device."""
    metadata = getattr(module, "parallel_metadata", None)
    if isinstance(device, str) and device.startswith("cuda") and metadata is not None:
        device = f"cuda:{metadata.local_device}"
    return nn.Module.to(module, device, *args, **kwargs)


def _to_cuda(module: nn.Module, *args, **kwargs):
    return _to_device(module, "cuda")
