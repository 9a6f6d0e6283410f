"""Greedy least-loaded optimizer-state sharding (reference: optim/zero/sharding.py,
credit there to fairscale OSS)."""
from typing import Dict, List


from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class OptimizerStateSharding:
    """Partition params across DP ranks, preserving param-group structure.

    Each param goes to the currently least-loaded rank (by numel).
    """

    def __init__(self, param_groups: List[Dict], parallel_context: ParallelContext,
                 parallel_mode: ParallelMode = ParallelMode.DATA):
        self.param_groups = param_groups
        self.parallel_context = parallel_context
        self.parallel_mode = parallel_mode

    def shard(self) -> List[List[Dict]]:
        world = self.parallel_context.get_world_size(self.parallel_mode)
        sizes = [0] * world
        # per rank, per group: {"params": [...], **group_opts}
        partitions: List[List[Dict]] = [
            [{k: v for k, v in g.items() if k != "params"} | {"params": []}
             for g in self.param_groups]
            for _ in range(world)
        ]
        for gi, group in enumerate(self.param_groups):
            for p in group["params"]:
                rank = sizes.index(min(sizes))
                partitions[rank][gi]["params"].append(p)
                sizes[rank] += p.numel()
        return partitions
