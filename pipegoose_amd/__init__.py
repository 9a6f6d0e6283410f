"""pipegoose_amd: an MI355X-native 4D-parallel training framework.

Capabilities of xrsrke/pipegoose (ParallelContext, TensorParallel,
DataParallel, PipelineParallel, ExpertParallel, ZeRO-1 DistributedOptimizer,
HF-transformers drop-in parallelize()), redesigned for MI355X:
PyTorch-ROCm autograd + hand-written CDNA4 (gfx950) HIP kernels for the hot
ops + RCCL-over-xGMI collectives.
"""
__version__ = "0.1.0"

from pipegoose_amd.distributed import ParallelContext, ParallelMode

__all__ = ["ParallelContext", "ParallelMode", "__version__"]
