from pipegoose_amd.nn.tensor_parallel.embedding import ParallelEmbedding
from pipegoose_amd.nn.tensor_parallel.layer_norm import LayerNorm
from pipegoose_amd.nn.tensor_parallel.linear import ColumnParallelLinear, RowParallelLinear
from pipegoose_amd.nn.tensor_parallel.loss import VocabParallelCrossEntropy
from pipegoose_amd.nn.tensor_parallel.tensor_parallel import TensorParallel

__all__ = [
    "TensorParallel",
    "ColumnParallelLinear",
    "RowParallelLinear",
    "ParallelEmbedding",
    "LayerNorm",
    "VocabParallelCrossEntropy",
]
