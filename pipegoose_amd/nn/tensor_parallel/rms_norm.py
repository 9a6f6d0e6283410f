"""RMSNorm for TP/SP models — sibling of layer_norm.py.

GPU: fused CDNA4 kernel (ops/csrc/rms_norm.hip); CPU: fp32 eager oracle.
Same sequence-parallel treatment as LayerNorm: under SP the (replicated)
weight grad is partial over local tokens, fixed in-graph via _Broadcast.

Beyond-reference (the reference had no RMSNorm/llama support).
"""
import torch
from torch import nn

from pipegoose_amd.ops.rms_norm import fused_rms_norm


class RMSNorm(nn.Module):
    def __init__(self, hidden_size: int, eps: float = 1e-6,
                 sequence_parallel: bool = False, parallel_context=None):
        super().__init__()
        self.eps = eps
        self.sequence_parallel = sequence_parallel and parallel_context is not None
        self.parallel_context = parallel_context
        self.weight = nn.Parameter(torch.ones(hidden_size))

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        weight = self.weight
        if getattr(self, "sequence_parallel", False):
            from pipegoose_amd.nn.tensor_parallel._functional import (
                broadcast_to_tensor_group)
            weight = broadcast_to_tensor_group(weight, self.parallel_context)
        return fused_rms_norm(input, weight, self.eps)
