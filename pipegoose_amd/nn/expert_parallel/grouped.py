"""Grouped expert GEMM: all local experts' FFNs as ONE batched MFMA GEMM.

The per-expert loop launches E small GEMMs that each underfill the 256-CU
chip; here the expert-sorted token buffer (the all-to-all dispatch already
produces it) is padded per expert to a common row count and the whole bank
runs as two ``torch.bmm`` calls ([E, maxN, H] x [E, H, I]) — hipBLASLt
batched MFMA kernels — with the activation between.  North-star item
"grouped expert GEMM" (BASELINE.json): batching beats hand-tiling here
because hipBLASLt's batched kernels already reach the MFMA roofline for
these shapes; a hand kernel must match it or fall back (guide §5 rule).

MI355X-native replacement for the reference's per-expert Python loop
(nn/expert_parallel/experts.py:41-82; SURVEY §2.5 'grouped-GEMM HIP kernel').
"""
from typing import List

import torch
from torch import nn


_SUPPORTED_ACTS = (nn.GELU, nn.SiLU, nn.ReLU, nn.Tanh)


def match_grouped_mlp(experts: nn.ModuleList):
    """If every expert is Sequential(Linear, act, Linear) with identical
    shapes, return (w1s, b1s, w2s, b2s, act) views; else None."""
    w1, b1, w2, b2 = [], [], [], []
    act = None
    for e in experts:
        if not isinstance(e, nn.Sequential) or len(e) != 3:
            return None
        lin1, a, lin2 = e[0], e[1], e[2]
        if not (isinstance(lin1, nn.Linear) and isinstance(lin2, nn.Linear)
                and isinstance(a, _SUPPORTED_ACTS)):
            return None
        if act is None:
            act = a
        elif type(a) is not type(act):
            return None
        w1.append(lin1.weight); b1.append(lin1.bias)
        w2.append(lin2.weight); b2.append(lin2.bias)
    if any(b is None for b in b1) != all(b is None for b in b1):
        return None
    return w1, b1, w2, b2, act


class _GroupedLinear(torch.autograd.Function):
    """Batched y = x @ W^T with CONTIGUOUS-operand bmms on both passes.

    torch.bmm with a transposed (batch-strided) operand mem-faults in the
    ROCm GEMM backend at the 1b7-MoE shapes (bisected: tools/moe_repro.py)
    — and autograd's built-in bmm backward issues exactly such transposed
    views, so the workaround must own the backward too: dx = dy·W and
    dW = dy^T·x with explicit contiguous copies.
    """

    @staticmethod
    def forward(ctx, x, W):              # x [E, M, K], W [E, N, K]
        ctx.save_for_backward(x, W)
        return torch.bmm(x.contiguous(), W.transpose(1, 2).contiguous())

    @staticmethod
    def backward(ctx, dy):               # dy [E, M, N]
        x, W = ctx.saved_tensors
        dy = dy.contiguous()
        dx = torch.bmm(dy, W.contiguous())                    # [E, M, K]
        dW = torch.bmm(dy.transpose(1, 2).contiguous(), x.contiguous())
        return dx, dW


def grouped_mlp_forward(tokens: torch.Tensor, counts: List[int],
                        w1: List[torch.Tensor], b1, w2: List[torch.Tensor],
                        b2, act: nn.Module) -> torch.Tensor:
    """tokens: [N, H] sorted by local expert with ``counts[i]`` rows each.
    Returns [N, H_out] in the same order.  Differentiable (stack + bmm)."""
    E = len(counts)
    H = tokens.size(-1)
    max_n = max(max(counts), 1)
    dev, dt = tokens.device, tokens.dtype

    padded = torch.zeros(E, max_n, H, device=dev, dtype=dt)
    start = 0
    for i, c in enumerate(counts):
        if c:
            padded[i, :c] = tokens[start:start + c]
        start += c

    W1 = torch.stack(w1).to(dt)          # [E, I, H]
    W2 = torch.stack(w2).to(dt)          # [E, H_out, I]
    h = _GroupedLinear.apply(padded, W1)
    if b1[0] is not None:
        h = h + torch.stack(b1).to(dt).unsqueeze(1)
    h = act(h)
    out = _GroupedLinear.apply(h, W2)
    if b2[0] is not None:
        out = out + torch.stack(b2).to(dt).unsqueeze(1)

    segs = [out[i, :c] for i, c in enumerate(counts) if c]
    if not segs:
        return tokens[:0]
    return torch.cat(segs, dim=0)
