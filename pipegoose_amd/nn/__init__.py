from pipegoose_amd.nn.data_parallel.data_parallel import DataParallel
from pipegoose_amd.nn.expert_parallel.expert_parallel import ExpertParallel
from pipegoose_amd.nn.tensor_parallel.tensor_parallel import TensorParallel

__all__ = ["DataParallel", "TensorParallel", "ExpertParallel", "PipelineParallel"]


def __getattr__(name):
    # PipelineParallel imported lazily (pulls in the partitioner/fx machinery)
    if name == "PipelineParallel":
        from pipegoose_amd.nn.pipeline_parallel.pipeline_parallel import PipelineParallel
        return PipelineParallel
    raise AttributeError(name)
