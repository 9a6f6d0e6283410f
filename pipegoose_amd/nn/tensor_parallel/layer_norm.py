"""LayerNorm for TP models (full-width; reference: nn/tensor_parallel/layer_norm.py).

On GPU this dispatches to the hand-written CDNA4 fused LayerNorm kernel
(pipegoose_amd.ops.layer_norm — one-pass wave-reduction fwd, fused bwd);
on CPU it falls back to torch.nn.functional.layer_norm.
"""
import torch
from torch import nn

from pipegoose_amd.ops.layer_norm import fused_add_layer_norm, fused_layer_norm


class LayerNorm(nn.Module):
    def __init__(self, normalized_shape, eps: float = 1e-5,
                 sequence_parallel: bool = False, parallel_context=None):
        super().__init__()
        if isinstance(normalized_shape, int):
            normalized_shape = (normalized_shape,)
        self.normalized_shape = tuple(normalized_shape)
        self.eps = eps
        self.parallel_context = parallel_context
        # SP: this norm sees only the local [B, S/tp, H] shard, so its
        # (replicated) weight/bias grads are partial sums over local tokens;
        # routing the params through _Broadcast (identity fwd, all-reduce bwd
        # over TENSOR) restores the full gradient per backward — correct under
        # microbatch accumulation too.
        self.sequence_parallel = sequence_parallel and parallel_context is not None
        self.weight = nn.Parameter(torch.ones(self.normalized_shape))
        self.bias = nn.Parameter(torch.zeros(self.normalized_shape))

    def _params(self):
        weight, bias = self.weight, self.bias
        if getattr(self, "sequence_parallel", False):
            from pipegoose_amd.nn.tensor_parallel._functional import (
                broadcast_to_tensor_group)
            weight = broadcast_to_tensor_group(weight, self.parallel_context)
            bias = broadcast_to_tensor_group(bias, self.parallel_context)
        return weight, bias

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        weight, bias = self._params()
        return fused_layer_norm(input, self.normalized_shape, weight, bias,
                                self.eps)

    def forward_with_residual(self, input: torch.Tensor, residual):
        """(LN(input + residual), input + residual) — residual add fused into
        the norm's HBM pass on GPU; residual may be None."""
        weight, bias = self._params()
        return fused_add_layer_norm(input, residual, self.normalized_shape,
                                    weight, bias, self.eps)
