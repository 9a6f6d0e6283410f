"""Fused LayerNorm: one-pass mean/var with 64-wide wave reductions on gfx950.

CPU path: torch.nn.functional.layer_norm (the numerics oracle the GPU kernel
is tested against, tests/ops/test_layer_norm.py).
"""
import os

import torch
import torch.nn.functional as TF

from pipegoose_amd.ops import get_extension


class _FusedLayerNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, bias, eps):
        ext = get_extension(required=True)
        y, mean, rstd = ext.layer_norm_fwd(input, weight, bias, eps)
        ctx.save_for_backward(input, weight, mean, rstd)
        return y

    @staticmethod
    def backward(ctx, grad_out):
        ext = get_extension(required=True)
        input, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.layer_norm_bwd(grad_out.contiguous(), input, weight, mean, rstd)
        return dx, dw, db, None


def fused_layer_norm(input, normalized_shape, weight, bias, eps=1e-5):
    if input.is_cuda and os.environ.get("PIPEGOOSE_DISABLE_EXT") != "1":
        return _FusedLayerNorm.apply(input.contiguous(), weight, bias, eps)
    return TF.layer_norm(input, normalized_shape, weight, bias, eps)
