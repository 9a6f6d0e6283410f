"""Megatron-style 1D tensor-parallel linear layers.

Reference parity: nn/tensor_parallel/linear.py:40-82.  The local GEMM runs on
hipBLASLt via torch.matmul (library GEMM); fused bias(+GeLU) epilogues use the
hand-written CDNA4 kernels in pipegoose_amd.ops when on GPU.

Fixes vs reference: ``bias=False`` is supported (reference crashed,
linear.py:44).
"""
import torch
from torch import nn
import torch.nn.functional as TF

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.tensor_parallel._functional import (
    all_gather_sequence,
    broadcast_to_tensor_group,
    gather_to_tensor_group,
    reduce_scatter_sequence,
    reduce_to_tensor_group,
    scatter_to_tensor_group,
)


def _local_linear(mod, input, bias):
    """The local GEMM of a TP linear: bf16 hipBLASLt by default; fp8
    (dynamic-scale e4m3, ops/fp8.py) when the layer carries the ``fp8``
    flag set by ``convert_linear_to_fp8`` — collectives around it are
    unchanged."""
    if getattr(mod, "fp8", False):
        from pipegoose_amd.ops.fp8 import fp8_linear
        return fp8_linear(input, mod.weight, bias)
    return TF.linear(input, mod.weight, bias)


class ColumnParallelLinear(nn.Module):
    """Y = XW^T + b with W split along output dim: each rank computes a slice
    of the output features; optionally all-gathered along the last dim."""

    def __init__(
        self,
        in_features: int,
        out_features: int,
        bias: bool = True,
        gather_output: bool = False,
        sequence_parallel: bool = False,
        parallel_context: ParallelContext = None,
    ):
        super().__init__()
        world = parallel_context.get_world_size(ParallelMode.TENSOR)
        assert out_features % world == 0
        self.in_features = in_features
        self.out_features = out_features // world
        self.gather_output = gather_output
        # Megatron-SP: input arrives sequence-sharded [B, S/tp, H]; forward
        # all-gathers S, backward reduce-scatters the input grad (replacing
        # _Broadcast's all-reduce — same wire bytes, sharded activations).
        self.sequence_parallel = sequence_parallel and world > 1
        self.parallel_context = parallel_context
        self.weight = nn.Parameter(torch.empty(self.out_features, in_features))
        if bias:
            self.bias = nn.Parameter(torch.empty(self.out_features))
        else:
            self.register_parameter("bias", None)

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        if getattr(self, "sequence_parallel", False):
            input = all_gather_sequence(input, self.parallel_context, dim=1)
        else:
            input = broadcast_to_tensor_group(input, self.parallel_context)
        output = _local_linear(self, input, self.bias)
        if self.gather_output:
            output = gather_to_tensor_group(output, dim=-1,
                                            parallel_context=self.parallel_context)
        return output


class RowParallelLinear(nn.Module):
    """Y = XW^T + b with W split along input dim: input is scattered along the
    last dim, partial products are all-reduced, bias added once (unsliced)."""

    def __init__(
        self,
        in_features: int,
        out_features: int,
        bias: bool = True,
        sequence_parallel: bool = False,
        parallel_context: ParallelContext = None,
    ):
        super().__init__()
        world = parallel_context.get_world_size(ParallelMode.TENSOR)
        assert in_features % world == 0
        self.in_features = in_features // world
        self.out_features = out_features
        # Megatron-SP: partial outputs reduce-scatter along S instead of
        # all-reduce; result is the complete [B, S/tp, H] shard.
        self.sequence_parallel = sequence_parallel and world > 1
        self.parallel_context = parallel_context
        self.weight = nn.Parameter(torch.empty(out_features, self.in_features))
        if bias:
            self.bias = nn.Parameter(torch.empty(out_features))
        else:
            self.register_parameter("bias", None)

    def forward(self, input: torch.Tensor) -> torch.Tensor:
        tp = self.parallel_context.get_world_size(ParallelMode.TENSOR)
        if tp == 1:
            # no combine needed: let hipBLASLt fuse the bias epilogue
            return _local_linear(self, input, self.bias)
        if input.size(-1) == self.in_features * tp:
            input = scatter_to_tensor_group(input, dim=-1,
                                            parallel_context=self.parallel_context)
        output = _local_linear(self, input, None)
        if getattr(self, "sequence_parallel", False):
            output = reduce_scatter_sequence(output, self.parallel_context, dim=1)
            if self.bias is not None:
                # replicated bias sees only the S-shard: restore the full
                # grad with an identity-fwd / all-reduce-bwd on the param
                output = output + broadcast_to_tensor_group(
                    self.bias, self.parallel_context)
            return output
        output = reduce_to_tensor_group(output, self.parallel_context)
        if self.bias is not None:
            output = output + self.bias
        return output
