"""Experimental fp8 (e4m3/e5m2) linear path for MI355X.

CDNA4's matrix cores run fp8 at 2x the bf16 rate; measured through
hipBLASLt (`torch._scaled_mm`) on the TP-linear bench shapes this build
delivers 1.8-3.2 PF vs 1.0-1.6 PF bf16 (profiles/fp8_probe — up to
2.04x).  This module provides:

- dynamic per-tensor quantization helpers (e4m3 for activations/weights,
  e5m2 for gradients — the standard recipe: gradients need e5m2's range,
  activations e4m3's precision),
- ``_Fp8LinearFn``: full fwd+bwd linear where all three GEMMs (fwd,
  dgrad, wgrad) run in fp8 with bf16 accumulate-out,
- ``Fp8Linear`` (drop-in ``nn.Linear`` subclass) and
  ``convert_linear_to_fp8`` (in-place class swap, same surgery idiom as
  the TP parallelizer),

Quantization is dynamic (abs-max per tensor per call) — no calibration
state to manage; the cast+amax overhead is why only large GEMMs benefit
(see tools/fp8_linear_bench.py; decode-sized GEMMs stay bf16).

EXPERIMENTAL: the framework's benchmarks and default training path stay
bf16 (BASELINE.json's dtype).  This is the measured on-ramp for an fp8
recipe, not a silently-enabled precision drop.  Beyond-reference
capability (the reference is fp32/fp16-only CUDA).
"""
from typing import Iterable, Optional

import torch
import torch.nn as nn

E4M3_MAX = 448.0
E5M2_MAX = 57344.0


def quantize(t: torch.Tensor, dtype: torch.dtype):
    """Per-tensor dynamic scale: returns (q, scale) with q ≈ t / scale.

    ``scale`` is the DEQUANT factor (torch._scaled_mm convention:
    out = (scale_a * a8) @ (scale_b * b8)).  On GPU the HIP kernel does
    it in one amax pass + one cast pass (csrc/fp8_quant.hip) — the eager
    route below is 4-5 fp32 passes and is kept as the CPU oracle."""
    if t.is_cuda and t.dtype == torch.bfloat16:
        from pipegoose_amd.ops import get_extension
        ext = get_extension(required=False)
        if ext is not None and hasattr(ext, "fp8_quant"):
            q, sc = ext.fp8_quant(t.contiguous(),
                                  dtype == torch.float8_e5m2)
            return q, sc[0]
    max_val = E4M3_MAX if dtype == torch.float8_e4m3fn else E5M2_MAX
    amax = t.abs().amax().float().clamp(min=1e-12)
    scale = amax / max_val
    q = (t.float() / scale).clamp(-max_val, max_val).to(dtype)
    return q, scale


def _transpose_q(q: torch.Tensor) -> torch.Tensor:
    """Contiguous byte transpose of a 2D fp8 tensor (LDS-tiled kernel on
    GPU; the eager fallback is a strided copy)."""
    if q.is_cuda:
        from pipegoose_amd.ops import get_extension
        ext = get_extension(required=False)
        if ext is not None and hasattr(ext, "fp8_transpose"):
            return ext.fp8_transpose(q)
    return q.t().contiguous()


def _quant_dual(t: torch.Tensor, dtype: torch.dtype):
    """quantize() that also returns the transposed fp8 image: one fused
    kernel on GPU, quantize+transpose fallback elsewhere."""
    if t.is_cuda and t.dtype == torch.bfloat16 and t.dim() == 2 \
            and t.size(1) % 4 == 0:
        from pipegoose_amd.ops import get_extension
        ext = get_extension(required=False)
        if ext is not None and hasattr(ext, "fp8_quant_dual"):
            q, qt, sc = ext.fp8_quant_dual(t.contiguous(),
                                           dtype == torch.float8_e5m2)
            return q, qt, sc[0]
    q, s = quantize(t, dtype)
    return q, _transpose_q(q), s


def dequantize(q: torch.Tensor, scale: torch.Tensor,
               dtype: torch.dtype = torch.float32) -> torch.Tensor:
    return q.to(torch.float32).mul(scale).to(dtype)


def _scaled_mm(a8, b8, sa, sb, out_dtype):
    # a8 row-major [M,K]; b8 must be column-major [K,N]
    return torch._scaled_mm(a8, b8, scale_a=sa, scale_b=sb,
                            out_dtype=out_dtype)


class _Fp8LinearFn(torch.autograd.Function):
    """y = x @ W^T + b with all three GEMMs in fp8.

    fwd:   x(e4m3) [M,K] @ W(e4m3)^T       (W [N,K] row-major -> col-major view)
    dgrad: dy(e5m2) [M,N] @ W(e4m3) [N,K]  (needs a col-major W copy)
    wgrad: dy^T(e5m2) [N,M] @ x(e4m3) [M,K] (row/col-major copies)
    """

    @staticmethod
    def forward(ctx, x, weight, bias):
        shp = x.shape
        x2 = x.reshape(-1, shp[-1])
        # dual quantize: the transposed images the backward GEMM layouts
        # need are produced in the same read (csrc cast_dual_kernel) —
        # no separate byte-transpose pass at backward time
        x8, x8t, sx = _quant_dual(x2, torch.float8_e4m3fn)
        w8, w8t, sw = _quant_dual(weight, torch.float8_e4m3fn)
        y = _scaled_mm(x8, w8.t(), sx, sw, x.dtype)
        if bias is not None:
            y = y + bias
        ctx.save_for_backward(x8t, sx, w8t, sw)
        ctx.has_bias = bias is not None
        ctx.in_shape = shp
        return y.reshape(*shp[:-1], weight.size(0))

    @staticmethod
    def backward(ctx, dy):
        x8t, sx, w8t, sw = ctx.saved_tensors
        dy2 = dy.reshape(-1, dy.size(-1))
        dy8, dy8t, sdy = _quant_dual(dy2, torch.float8_e5m2)
        # dgrad: dy [M,N] @ W [N,K] — scaled_mm wants mat2 column-major,
        # i.e. the [K,N] row-major transposed image viewed back
        dx = _scaled_mm(dy8, w8t.t(), sdy, sw, dy.dtype)
        # wgrad: dy^T [N,M] @ x [M,K] (col-major via the [K,M] image)
        dw = _scaled_mm(dy8t, x8t.t(), sdy, sx, dy.dtype)
        db = dy2.sum(0) if ctx.has_bias else None
        return dx.reshape(ctx.in_shape), dw, db


def _eligible(x: torch.Tensor, weight: torch.Tensor) -> bool:
    return (x.is_cuda and x.size(-1) % 16 == 0
            and weight.size(0) % 16 == 0
            and (x.numel() // x.size(-1)) % 16 == 0)


def fp8_linear(x: torch.Tensor, weight: torch.Tensor,
               bias: Optional[torch.Tensor] = None) -> torch.Tensor:
    """F.linear that runs in fp8 when eligible (CUDA, dims % 16)."""
    if _eligible(x, weight):
        return _Fp8LinearFn.apply(x, weight, bias)
    return torch.nn.functional.linear(x, weight, bias)


class Fp8Linear(nn.Linear):
    """nn.Linear whose GEMMs run in fp8 on CUDA (bf16 elsewhere / when
    shapes are not fp8-eligible: hipBLASLt needs K,N % 16 == 0)."""

    def forward(self, x):
        if _eligible(x, self.weight):
            return _Fp8LinearFn.apply(x, self.weight, self.bias)
        return super().forward(x)


def convert_linear_to_fp8(module: nn.Module,
                          names: Optional[Iterable[str]] = None) -> int:
    """Route a model's linears through fp8, in place.  Plain ``nn.Linear``
    children get the class swap (same surgery idiom as the TP
    parallelizer); the TP layers (Column/RowParallelLinear — incl. the
    fused_bias_gelu path) get an ``fp8`` flag their forward consults, so
    the surrounding collectives/epilogues are untouched.  ``names``
    restricts to submodule-name suffixes (e.g. ["dense_h_to_4h",
    "dense_4h_to_h"] = the MLP GEMMs); None converts everything.
    Returns the number converted."""
    from pipegoose_amd.nn.tensor_parallel.linear import (ColumnParallelLinear,
                                                         RowParallelLinear)
    n = 0
    for name, child in module.named_modules():
        if names is not None and not any(name.endswith(s) for s in names):
            continue
        if type(child) is nn.Linear:
            child.__class__ = Fp8Linear
            n += 1
        elif isinstance(child, (ColumnParallelLinear, RowParallelLinear)):
            child.fp8 = True
            n += 1
    return n


def revert_fp8(module: nn.Module) -> int:
    """Undo convert_linear_to_fp8."""
    n = 0
    for child in module.modules():
        if type(child) is Fp8Linear:
            child.__class__ = nn.Linear
            n += 1
        elif getattr(child, "fp8", False):
            child.fp8 = False
            n += 1
    return n
