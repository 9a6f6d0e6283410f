"""Match-or-fallback dispatch for the hand-written MFMA GEMM.

The hand kernel (csrc/gemm.hip, `gemm_bt`) computes the TP-linear forward
shape C = A·B^T with fused bias(+GeLU).  Round-2 measurement
(profiles/gemm_match_or_fallback_r02.md) shows hipBLASLt ahead on every
bench shape, so `WIN_SHAPES` is empty and `linear_forward` routes to
torch.nn.functional.linear (hipBLASLt) unless a shape is listed or
PG_HAND_GEMM=1 forces the hand kernel (for measurement).
"""
import os

import torch
import torch.nn.functional as TF

from pipegoose_amd.ops import get_extension

# (M, N, K) tuples where gemm_bt measured >= hipBLASLt.  Empty by
# measurement (see profiles/gemm_match_or_fallback_r02.md), not omission.
WIN_SHAPES = set()


def _use_hand(M: int, N: int, K: int) -> bool:
    if M % 128 or N % 128 or K % 64:
        return False
    if os.environ.get("PG_HAND_GEMM") == "1":
        return True
    return (M, N, K) in WIN_SHAPES


def hand_linear(x: torch.Tensor, weight: torch.Tensor,
                bias: torch.Tensor = None, gelu: bool = False) -> torch.Tensor:
    """y = x @ W^T (+bias)(+gelu) through the hand MFMA kernel; x any
    leading shape with contiguous 2D view."""
    ext = get_extension(required=True)
    lead = x.shape[:-1]
    x2 = x.reshape(-1, x.size(-1)).contiguous()
    out = ext.gemm_bt(x2, weight.contiguous(), bias, gelu)
    return out.reshape(*lead, weight.size(0))


def linear_forward(x: torch.Tensor, weight: torch.Tensor,
                   bias: torch.Tensor = None) -> torch.Tensor:
    """TP-linear local GEMM with per-shape match-or-fallback routing."""
    if (x.is_cuda and x.dtype == torch.bfloat16
            and weight.dtype == torch.bfloat16 and not torch.is_grad_enabled()):
        M = x.numel() // x.size(-1)
        if _use_hand(M, weight.size(0), x.size(-1)):
            return hand_linear(x, weight, bias)
    return TF.linear(x, weight, bias)
