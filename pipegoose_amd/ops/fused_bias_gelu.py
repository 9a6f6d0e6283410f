"""Fused bias + GeLU epilogue.

On GPU: GEMM via hipBLASLt (torch.matmul) + one fused HIP kernel applying
bias+GeLU (tanh approx) in a single HBM pass, with a fused dgelu backward.
CPU path: eager bias + torch GELU (the numerics oracle).

MI355X-native fusion of the reference's separate bias-add + GELU modules
('fused bias-GeLU for mlp.dense_h_to_4h').
"""
import os

import torch
import torch.nn.functional as TF

from pipegoose_amd.ops import get_extension


class _BiasGelu(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, bias):
        ext = get_extension(required=True)
        out = ext.bias_gelu_fwd(input, bias)
        ctx.save_for_backward(input, bias)
        return out

    @staticmethod
    def backward(ctx, grad_out):
        ext = get_extension(required=True)
        input, bias = ctx.saved_tensors
        # bias grad accumulated in the same HBM pass as dx (fp32 atomics)
        grad_in, grad_bias = ext.bias_gelu_bwd(grad_out.contiguous(), input, bias)
        return grad_in, grad_bias


def bias_gelu(input: torch.Tensor, bias: torch.Tensor) -> torch.Tensor:
    if input.is_cuda and os.environ.get("PIPEGOOSE_DISABLE_EXT") != "1":
        return _BiasGelu.apply(input.contiguous(), bias)
    return TF.gelu(input + bias, approximate="tanh")


def fused_bias_gelu(column_linear, hidden: torch.Tensor) -> torch.Tensor:
    """Run a ColumnParallelLinear with its bias+GeLU fused into the epilogue."""
    from pipegoose_amd.nn.tensor_parallel._functional import (
        all_gather_sequence, broadcast_to_tensor_group)
    if getattr(column_linear, "sequence_parallel", False):
        hidden = all_gather_sequence(hidden, column_linear.parallel_context, dim=1)
    else:
        hidden = broadcast_to_tensor_group(hidden, column_linear.parallel_context)
    if getattr(column_linear, "fp8", False):
        from pipegoose_amd.ops.fp8 import fp8_linear
        x = fp8_linear(hidden, column_linear.weight)      # fp8 GEMM, no bias
    else:
        x = TF.linear(hidden, column_linear.weight)  # hipBLASLt GEMM, no bias
    if column_linear.bias is not None:
        return bias_gelu(x, column_linear.bias)
    return TF.gelu(x, approximate="tanh")
