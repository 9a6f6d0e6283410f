from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode

__all__ = ["ParallelContext", "ParallelMode"]
