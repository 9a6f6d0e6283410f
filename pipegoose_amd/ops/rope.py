"""Rotary position embedding, half-split (NeoX/LLaMA) convention.

GPU: in-kernel cos/sin from the base frequency (no [S, D] tables in HBM);
backward is the inverse rotation.  CPU oracle: explicit fp32 rotation.

Beyond-reference (the reference had no RoPE/llama support); SURVEY §2.7 item 1
names RoPE as a required native kernel.
"""
import os

import torch

from pipegoose_amd.ops import get_extension


class _Rope(torch.autograd.Function):
    @staticmethod
    def forward(ctx, x, theta_base, pos_offset):
        ext = get_extension(required=True)
        ctx.theta_base = theta_base
        ctx.pos_offset = pos_offset
        return ext.rope_apply(x, theta_base, False, pos_offset)

    @staticmethod
    def backward(ctx, dy):
        ext = get_extension(required=True)
        return ext.rope_apply(dy.contiguous(), ctx.theta_base, True,
                              ctx.pos_offset), None, None


def _rope_ref(x: torch.Tensor, theta_base: float,
              pos_offset: int = 0) -> torch.Tensor:
    B, H, S, D = x.shape
    half = D // 2
    d = torch.arange(half, device=x.device, dtype=torch.float32)
    freqs = theta_base ** (-2.0 * d / D)
    pos = torch.arange(pos_offset, pos_offset + S, device=x.device,
                       dtype=torch.float32)
    ang = pos[:, None] * freqs
    cos, sin = ang.cos(), ang.sin()          # [S, D/2]
    x1, x2 = x[..., :half].float(), x[..., half:].float()
    y1 = x1 * cos - x2 * sin
    y2 = x2 * cos + x1 * sin
    return torch.cat([y1, y2], dim=-1).to(x.dtype)


def apply_rope(x: torch.Tensor, theta_base: float = 10000.0,
               pos_offset: int = 0) -> torch.Tensor:
    """x: [B, H, S, D] — rotate at absolute positions pos_offset..+S-1."""
    if x.is_cuda and os.environ.get("PIPEGOOSE_DISABLE_EXT") != "1":
        return _Rope.apply(x.contiguous(), theta_base, pos_offset)
    return _rope_ref(x, theta_base, pos_offset)


def rope_at_position(x: torch.Tensor, theta_base: float,
                     pos_t: torch.Tensor) -> torch.Tensor:
    """x: [B, H, 1, D], pos_t: int64 device tensor [1] — rotate the single
    decode step at a DATA-carried position (hipGraph-capturable: no host
    value enters; same half-split math as _rope_ref)."""
    half = x.size(-1) // 2
    d = torch.arange(half, device=x.device, dtype=torch.float32)
    freqs = theta_base ** (-2.0 * d / x.size(-1))
    ang = pos_t.float()[:, None] * freqs          # [1, half]
    cos, sin = ang.cos(), ang.sin()
    x1, x2 = x[..., :half].float(), x[..., half:].float()
    y1 = x1 * cos - x2 * sin
    y2 = x2 * cos + x1 * sin
    return torch.cat([y1, y2], dim=-1).to(x.dtype)
