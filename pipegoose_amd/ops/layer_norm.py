"""Fused LayerNorm: one-pass mean/var with 64-wide wave reductions on gfx950.

CPU path: torch.nn.functional.layer_norm (the numerics oracle the GPU kernel
is tested against, tests/ops/test_layer_norm.py).

MI355X-native kernel wrapper, no reference counterpart (the reference used
plain nn.LayerNorm — nn/tensor_parallel/layer_norm.py:23-25).
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


class _FusedAddLayerNorm(torch.autograd.Function):
    """y, s = LN(x + r), (x + r) in one HBM pass — the residual add that would
    otherwise be its own elementwise kernel is folded into the norm's read.
    Backward: d(x) = d(r) = LN-dx(dy) + ds (both residual branches see the
    same gradient)."""

    @staticmethod
    def forward(ctx, input, residual, weight, bias, eps):
        ext = get_extension(required=True)
        y, s, mean, rstd = ext.layer_norm_res_fwd(input, residual, weight, bias, eps)
        ctx.save_for_backward(s, weight, mean, rstd)
        return y, s

    @staticmethod
    def backward(ctx, grad_y, grad_s):
        ext = get_extension(required=True)
        s, weight, mean, rstd = ctx.saved_tensors
        dx, dw, db = ext.layer_norm_bwd(grad_y.contiguous(), s, weight, mean, rstd)
        if grad_s is not None:
            dx = dx + grad_s
        return dx, dx, dw, db, None


def fused_add_layer_norm(input, residual, normalized_shape, weight, bias,
                         eps=1e-5):
    """Returns (normed, summed): normed = LN(input + residual), summed = the
    residual stream to carry forward.  ``residual=None`` degrades to plain LN
    (returns (normed, input))."""
    if residual is None:
        return fused_layer_norm(input, normalized_shape, weight, bias, eps), input
    if input.is_cuda and os.environ.get("PIPEGOOSE_DISABLE_EXT") != "1":
        return _FusedAddLayerNorm.apply(input.contiguous(),
                                        residual.contiguous(), weight, bias, eps)
    s = input + residual
    return TF.layer_norm(s, normalized_shape, weight, bias, eps), s
