from pipegoose_amd.nn.pipeline_parallel.pipeline_parallel import PipelineParallel
from pipegoose_amd.nn.pipeline_parallel.scheduler import (
    GPipeScheduler,
    JobType,
    OneFOneBScheduler,
    Task,
)

__all__ = ["PipelineParallel", "GPipeScheduler", "OneFOneBScheduler", "Task", "JobType"]
