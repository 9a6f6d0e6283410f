"""Parallel modes: the axes of the 4D process grid.

Mirrors the reference's mode set (pipegoose/distributed/parallel_mode.py:4-12)
so user code keyed on modes ports directly.
"""
from enum import Enum


class ParallelMode(Enum):
    GLOBAL = "global"

    TENSOR = "tensor"
    PIPELINE = "pipeline"
    DATA = "data"

    # Data-parallel replication group for expert (MoE) parameters.  Experts are
    # sharded over the TENSOR axis, so their gradients reduce over a group with
    # the same layout as the DATA axis restricted to matching expert shards.
    EXPERT_DATA = "expert_data"

    # Context (sequence-block) parallelism: long sequences shard along S over
    # this group; ring attention rotates KV blocks around it (absent in the
    # reference — SURVEY.md §5 'Long-context').  Grid order is
    # [pipeline][data][context][tensor], tensor fastest.
    CONTEXT = "context"
