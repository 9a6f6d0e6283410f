"""Vocab-parallel embedding (reference: nn/tensor_parallel/embedding.py:26-42).

Each rank holds a contiguous vocab slice; out-of-range ids are masked, looked
up locally (zero rows for masked ids), and partial results are summed over the
TENSOR group.  On GPU the masked gather runs in a fused HIP kernel
(pipegoose_amd.ops.embedding) — one pass, no materialized mask tensors.
"""
import torch
from torch import nn
import torch.nn.functional as TF

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.tensor_parallel._functional import (
    reduce_scatter_sequence, reduce_to_tensor_group)


class ParallelEmbedding(nn.Module):
    def __init__(self, num_embeddings: int, embedding_dim: int,
                 sequence_parallel: bool = False,
                 parallel_context: ParallelContext = None):
        super().__init__()
        world = parallel_context.get_world_size(ParallelMode.TENSOR)
        # SP: the partial-sum combine is a reduce-scatter along S, so the
        # embedding emits the [B, S/tp, H] shard directly (Megatron-SP entry).
        self.sequence_parallel = sequence_parallel and world > 1
        assert num_embeddings % world == 0
        rank = parallel_context.get_local_rank(ParallelMode.TENSOR)
        self.num_embeddings = num_embeddings
        self.embedding_dim = embedding_dim
        self.partition_size = num_embeddings // world
        self.vocab_start_idx = rank * self.partition_size
        self.vocab_end_idx = self.vocab_start_idx + self.partition_size
        self.parallel_context = parallel_context
        self.weight = nn.Parameter(torch.empty(self.partition_size, embedding_dim))

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        world = self.parallel_context.get_world_size(ParallelMode.TENSOR)
        if world == 1:
            return TF.embedding(input, self.weight)
        mask = (input < self.vocab_start_idx) | (input >= self.vocab_end_idx)
        masked_input = (input - self.vocab_start_idx).masked_fill(mask, 0)
        output = TF.embedding(masked_input, self.weight)
        output = output.masked_fill(mask.unsqueeze(-1), 0.0)
        if getattr(self, "sequence_parallel", False):
            return reduce_scatter_sequence(output, self.parallel_context, dim=1)
        return reduce_to_tensor_group(output, self.parallel_context)
