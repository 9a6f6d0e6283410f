"""Process-group initializers for the 4D grid.

Rank layout (identical math to the reference so checkpoints/group semantics
match; see reference pipegoose/distributed/_initializers/initialize_*.py):

    world is ordered [pipeline][data][context][tensor], tensor fastest:
        global_rank = pp*(dp_size*cp_size*tp_size) + dp*(cp_size*tp_size)
                      + cp*tp_size + tp

  - TENSOR groups: contiguous blocks of ``tp_size`` ranks.
  - PIPELINE groups: strided, ``range(i, world, world // pp_size)``.
  - CONTEXT groups: within each (pp, dp) block, same tp offset across the
    cp blocks: ``range(base + tp_off, base + cp*tp, tp)``.
  - DATA groups: within each pipeline block, same (cp, tp) coordinate across
    dp: ``range(start + cp_off*tp + tp_off, end, cp*tp)``.
  - EXPERT_DATA groups: same rank layout as DATA groups (expert params are
    sharded over TENSOR, replicated over DATA).
    With cp == 1 (the default) every layout reduces to the reference's.

On MI355X a node has 8 GPUs fully connected over xGMI (7 point-to-point links
per GPU), so unlike NVSwitch there is no switch hop to co-locate around; the
contiguous-TP layout keeps TP collectives inside a node when scaling out.
"""
from dataclasses import dataclass
from typing import List, Optional

import torch.distributed as dist

from pipegoose_amd.distributed.parallel_mode import ParallelMode


@dataclass
class ProcessGroupResult:
    local_rank: int
    local_world_size: int
    ranks_in_group: List[int]
    process_group: Optional[dist.ProcessGroup]
    parallel_mode: ParallelMode


def _make_group(rank: int, ranks: List[int], mode: ParallelMode) -> ProcessGroupResult:
    # Every rank must call new_group for every group (collective contract).
    group = dist.new_group(ranks=ranks)
    if rank in ranks:
        return ProcessGroupResult(
            local_rank=ranks.index(rank),
            local_world_size=len(ranks),
            ranks_in_group=ranks,
            process_group=group,
            parallel_mode=mode,
        )
    return None


class ProcessGroupInitializer:
    def __init__(self, rank: int, world_size: int, tensor_parallel_size: int,
                 pipeline_parallel_size: int, data_parallel_size: int,
                 context_parallel_size: int = 1):
        self.rank = rank
        self.world_size = world_size
        self.tensor_parallel_size = tensor_parallel_size
        self.pipeline_parallel_size = pipeline_parallel_size
        self.data_parallel_size = data_parallel_size
        self.context_parallel_size = context_parallel_size

    def init_dist_group(self) -> ProcessGroupResult:
        raise NotImplementedError


class TensorParallelGroupInitializer(ProcessGroupInitializer):
    def init_dist_group(self) -> ProcessGroupResult:
        tp = self.tensor_parallel_size
        result = None
        for i in range(self.world_size // tp):
            ranks = list(range(i * tp, (i + 1) * tp))
            r = _make_group(self.rank, ranks, ParallelMode.TENSOR)
            result = r or result
        return result


class PipelineParallelGroupInitializer(ProcessGroupInitializer):
    def init_dist_group(self) -> ProcessGroupResult:
        pp = self.pipeline_parallel_size
        stride = self.world_size // pp
        result = None
        for i in range(stride):
            ranks = list(range(i, self.world_size, stride))
            r = _make_group(self.rank, ranks, ParallelMode.PIPELINE)
            result = r or result
        return result


class DataParallelGroupInitializer(ProcessGroupInitializer):
    def init_dist_group(self) -> ProcessGroupResult:
        tp = self.tensor_parallel_size
        pp = self.pipeline_parallel_size
        cp = self.context_parallel_size
        block = self.world_size // pp  # dp * cp * tp
        result = None
        for p in range(pp):
            start = p * block
            for j in range(cp * tp):
                ranks = list(range(start + j, start + block, cp * tp))
                r = _make_group(self.rank, ranks, ParallelMode.DATA)
                result = r or result
        return result


class ContextParallelGroupInitializer(ProcessGroupInitializer):
    """Sequence-block groups: rank order along the group == sequence order
    (ring attention rotates KV around exactly this ring)."""

    def init_dist_group(self) -> ProcessGroupResult:
        tp = self.tensor_parallel_size
        cp = self.context_parallel_size
        pp = self.pipeline_parallel_size
        block = self.world_size // pp
        result = None
        for p in range(pp):
            for d in range(block // (cp * tp)):
                base = p * block + d * cp * tp
                for j in range(tp):
                    ranks = list(range(base + j, base + cp * tp, tp))
                    r = _make_group(self.rank, ranks, ParallelMode.CONTEXT)
                    result = r or result
        return result


class ExpertDataParallelGroupInitializer(DataParallelGroupInitializer):
    """Replication group for expert parameters.

    Experts are sharded over the TENSOR axis; each expert shard is replicated
    across DATA-parallel replicas, so the expert grad-reduce group has the same
    rank layout as the DATA group (reference initialize_expert.py:10-44 uses the
    TP layout because its EP degree == TP degree; we key it off the DATA layout
    which is the group the gradients actually reduce over).
    """

    def init_dist_group(self) -> ProcessGroupResult:
        result = super().init_dist_group()
        if result is not None:
            result.parallel_mode = ParallelMode.EXPERT_DATA
        return result
