"""Fused SwiGLU: silu(gate) * up in one HBM pass (llama-family MLPs).

Beyond-reference (no llama family in the reference).
"""
import os

import torch
import torch.nn.functional as TF

from pipegoose_amd.ops import get_extension


class _SiluMul(torch.autograd.Function):
    @staticmethod
    def forward(ctx, gate, up):
        ext = get_extension(required=True)
        y = ext.silu_mul_fwd(gate, up)
        ctx.save_for_backward(gate, up)
        return y

    @staticmethod
    def backward(ctx, dy):
        ext = get_extension(required=True)
        gate, up = ctx.saved_tensors
        dg, du = ext.silu_mul_bwd(dy.contiguous(), gate, up)
        return dg, du


def silu_mul(gate: torch.Tensor, up: torch.Tensor) -> torch.Tensor:
    if gate.is_cuda and os.environ.get("PIPEGOOSE_DISABLE_EXT") != "1" \
            and gate.numel() % 8 == 0:
        return _SiluMul.apply(gate.contiguous(), up.contiguous())
    return TF.silu(gate) * up
