"""LayerNorm for TP models (full-width; reference: nn/tensor_parallel/layer_norm.py).

On GPU this dispatches to the hand-written CDNA4 fused LayerNorm kernel
(pipegoose_amd.ops.layer_norm — one-pass wave-reduction fwd, fused bwd);
on CPU it falls back to torch.nn.functional.layer_norm.
"""
import torch
from torch import nn

from pipegoose_amd.ops.layer_norm import fused_layer_norm


class LayerNorm(nn.Module):
    def __init__(self, normalized_shape, eps: float = 1e-5, parallel_context=None):
        super().__init__()
        if isinstance(normalized_shape, int):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.parallel_context = parallel_context
        self.weight = nn.Parameter(torch.ones(self.normalized_shape))
        self.bias = nn.Parameter(torch.zeros(self.normalized_shape))

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        return fused_layer_norm(input, self.normalized_shape, self.weight, self.bias,
                                self.eps)
