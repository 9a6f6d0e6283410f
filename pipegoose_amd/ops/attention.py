"""Causal ALiBi attention dispatch.

GPU path: hand-written CDNA4 flash-attention kernel with the ALiBi bias
slope*(j-i) computed in-kernel from the per-head slopes (no [H,S,S] mask
tensor) — see csrc/attention.hip.  Until/unless the kernel supports a given
shape, falls back to torch sdpa with an explicit additive mask.
CPU path: sdpa math with the mask (numerics oracle).

No reference counterpart (the reference has no attention kernels at all);
this is MI355X-native.
"""
import os

import torch
import torch.nn.functional as TF

from pipegoose_amd.ops import get_extension


def _kernel_supported(q: torch.Tensor) -> bool:
    if not q.is_cuda or os.environ.get("PIPEGOOSE_DISABLE_EXT") == "1":
        return False
    D = q.size(-1)
    S = q.size(-2)
    ext = get_extension()
    if ext is None or not hasattr(ext, "attn_fwd"):
        return False
    return D in (64, 128) and S % 64 == 0 and q.dtype == torch.bfloat16


def _d_contig(t: torch.Tensor) -> torch.Tensor:
    # kernels take arbitrary [B,H,S,D] strides as long as d is contiguous —
    # the model's transposed views qualify, so no copies happen here
    if os.environ.get("PG_ATTN_CONTIG") == "1":
        return t.contiguous()
    return t if t.stride(-1) == 1 else t.contiguous()


class _AlibiFlashAttention(torch.autograd.Function):
    @staticmethod
    def forward(ctx, q, k, v, slopes, scale):
        ext = get_extension(required=True)
        o, lse = ext.attn_fwd(q, k, v, slopes, scale, 0)
        ctx.save_for_backward(q, k, v, o, lse, slopes)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        ext = get_extension(required=True)
        q, k, v, o, lse, slopes = ctx.saved_tensors
        dq, dk, dv = ext.attn_bwd(_d_contig(do), q, k, v, o, lse, slopes,
                                  ctx.scale, 0)
        return dq, dk, dv, None, None


class _AlibiFlashAttentionFused(torch.autograd.Function):
    """Attention straight off the fused QKV projection [B, S, H, 3, hd]:
    forward reads q/k/v as strided views (no copies); backward writes
    dq/dk/dv DIRECTLY into the slices of one d(fused) buffer — without this,
    autograd assembles the three slice-grads with zeros+adds over the full
    [B, S, H, 3, hd] tensor three times (~5% of a training step)."""

    @staticmethod
    def forward(ctx, fused, slopes, scale):
        ext = get_extension(required=True)
        q = fused[:, :, :, 0, :].permute(0, 2, 1, 3)
        k = fused[:, :, :, 1, :].permute(0, 2, 1, 3)
        v = fused[:, :, :, 2, :].permute(0, 2, 1, 3)
        o, lse = ext.attn_fwd(q, k, v, slopes, scale, 0)
        ctx.save_for_backward(fused, o, lse, slopes)
        ctx.scale = scale
        return o

    @staticmethod
    def backward(ctx, do):
        ext = get_extension(required=True)
        fused, o, lse, slopes = ctx.saved_tensors
        q = fused[:, :, :, 0, :].permute(0, 2, 1, 3)
        k = fused[:, :, :, 1, :].permute(0, 2, 1, 3)
        v = fused[:, :, :, 2, :].permute(0, 2, 1, 3)
        dfused = torch.empty_like(fused)
        ext.attn_bwd_into(_d_contig(do), q, k, v, o, lse, slopes, ctx.scale,
                          dfused[:, :, :, 0, :].permute(0, 2, 1, 3),
                          dfused[:, :, :, 1, :].permute(0, 2, 1, 3),
                          dfused[:, :, :, 2, :].permute(0, 2, 1, 3), 0)
        return dfused, None, None


def alibi_attention_qkv(fused, slopes, scale):
    """fused: [B, S, H_local, 3, head_dim] straight from the QKV projection.
    Returns o with logical shape [B, H, S, hd] (physically [B, S, H, hd]).
    Caller must have checked _kernel_supported on the q view."""
    return _AlibiFlashAttentionFused.apply(
        fused, slopes.to(device=fused.device, dtype=torch.float32), scale)


def alibi_attention(q, k, v, slopes, scale, mask_fallback=None):
    """q,k,v: [B, H, S, D] (any strides, d contiguous); slopes: [H] fp32;
    causal + alibi bias computed in-kernel."""
    if _kernel_supported(q):
        return _AlibiFlashAttention.apply(
            _d_contig(q), _d_contig(k), _d_contig(v),
            slopes.to(device=q.device, dtype=torch.float32), scale)
    bias = mask_fallback(q.size(-2), q.device, q.dtype)
    return TF.scaled_dot_product_attention(q, k, v, attn_mask=bias, scale=scale)
