"""Fused RMSNorm (CPU oracle: explicit fp32 rms computation).

Beyond-reference kernel (no RMSNorm anywhere in the reference).
"""
import os

import torch

from pipegoose_amd.ops import get_extension


class _FusedRMSNorm(torch.autograd.Function):
    @staticmethod
    def forward(ctx, input, weight, eps):
        ext = get_extension(required=True)
        y, rstd = ext.rms_norm_fwd(input, weight, eps)
        ctx.save_for_backward(input, weight, rstd)
        return y

    @staticmethod
    def backward(ctx, grad_out):
        ext = get_extension(required=True)
        input, weight, rstd = ctx.saved_tensors
        dx, dw = ext.rms_norm_bwd(grad_out.contiguous(), input, weight, rstd)
        return dx, dw, None


def fused_rms_norm(input: torch.Tensor, weight: torch.Tensor,
                   eps: float = 1e-6) -> torch.Tensor:
    if input.is_cuda and os.environ.get("PIPEGOOSE_DISABLE_EXT") != "1":
        return _FusedRMSNorm.apply(input.contiguous(), weight, eps)
    xf = input.float()
    rstd = torch.rsqrt(xf.pow(2).mean(-1, keepdim=True) + eps)
    return (xf * rstd).to(input.dtype) * weight
