"""Memory profiling for non-uniform pipeline partitioning.

Reference parity: pipegoose/partitioning/profile.py:19-49 (`ProfileByMemory`)
measured `torch.cuda.memory_allocated` deltas per top-level layer plus param
storage bytes — and was never wired into the partitioner.  Here the profile
runs over the SAME ModelStructure decomposition the partitioner cuts
(pre / blocks / post), so its output feeds `UniformPartitioner(sizes=...)`
directly; on a GPU box the deltas come from the HIP caching allocator
(torch.cuda on ROCm), on CPU we fall back to parameter+activation-estimate
bytes so the CPU test path stays meaningful.
"""
from typing import List, Optional

import torch
from torch import nn


def _param_bytes(m: nn.Module) -> int:
    return sum(p.numel() * p.element_size() for p in m.parameters())


class ProfileByMemory:
    """Measure per-block memory cost of one forward pass.

    Usage:
        sizes = ProfileByMemory(model, device).profile(sample_input)
        stages = UniformPartitioner(model, ctx, sizes=sizes).split()
    """

    def __init__(self, model: nn.Module, device: Optional[torch.device] = None):
        self.model = model
        self.device = device or next(model.parameters()).device

    def _structure(self):
        from pipegoose_amd.nn.pipeline_parallel.partitioner import (
            UniformPartitioner)
        return UniformPartitioner(self.model, parallel_context=None)._structure()

    @torch.no_grad()
    def profile(self, input_ids: torch.Tensor) -> List[int]:
        """Returns one cost (bytes) per transformer block: parameters +
        measured activation memory of running that block."""
        s = self._structure()
        use_cuda = self.device.type == "cuda" and torch.cuda.is_available()

        x = input_ids.to(self.device)
        for m in s.pre:
            x = m(x)

        sizes: List[int] = []
        for block in s.blocks:
            if use_cuda:
                torch.cuda.synchronize(self.device)
                torch.cuda.reset_peak_memory_stats(self.device)
                before = torch.cuda.memory_allocated(self.device)
                x = block(x)
                torch.cuda.synchronize(self.device)
                act = max(torch.cuda.max_memory_allocated(self.device) - before, 0)
            else:
                x = block(x)
                out = x[0] if isinstance(x, tuple) else x
                act = out.numel() * out.element_size()
            sizes.append(_param_bytes(block) + int(act))
        return sizes
