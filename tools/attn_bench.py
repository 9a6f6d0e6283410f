"""Microbenchmark: hand-written attn kernels vs the sdpa-with-mask fallback.

Usage (GPU box):  python tools/attn_bench.py
Prints us/call for fwd and fwd+bwd on the bench shapes.
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipegoose_amd.ops import get_extension  # noqa: E402


def bench(fn, iters=20, warmup=5):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e6


def run_shape(B, H, S, D):
    ext = get_extension(required=True)
    torch.manual_seed(0)
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    do = torch.randn_like(q)
    slopes = (torch.rand(H, device="cuda") * 0.5).float()
    scale = D ** -0.5

    t_fwd = bench(lambda: ext.attn_fwd(q, k, v, slopes, scale, 0))
    o, lse = ext.attn_fwd(q, k, v, slopes, scale, 0)
    t_bwd = bench(lambda: ext.attn_bwd(do, q, k, v, o, lse, slopes, scale, 0))

    # sdpa-with-mask fallback (what the model used before the kernel)
    pos = torch.arange(S, device="cuda")
    rel = (pos[None, :] - pos[:, None]).float()
    bias = (slopes[:, None, None] * rel[None]
            + torch.triu(torch.full((S, S), float("-inf"), device="cuda"), 1)[None]
            ).to(torch.bfloat16)[None]
    qg = q.clone().requires_grad_(True)
    kg = k.clone().requires_grad_(True)
    vg = v.clone().requires_grad_(True)

    def sdpa_fwd():
        return torch.nn.functional.scaled_dot_product_attention(
            q, k, v, attn_mask=bias, scale=scale)

    t_sdpa_fwd = bench(sdpa_fwd)

    def sdpa_fwdbwd():
        out = torch.nn.functional.scaled_dot_product_attention(
            qg, kg, vg, attn_mask=bias, scale=scale)
        out.backward(do)
        qg.grad = kg.grad = vg.grad = None

    t_sdpa_fb = bench(sdpa_fwdbwd)

    flops_fwd = 2 * B * H * S * S * D * 2 / 2  # causal half
    print(f"B{B} H{H} S{S} D{D}: "
          f"ours fwd {t_fwd:8.1f}us ({flops_fwd / t_fwd / 1e6:6.1f} TF)  "
          f"bwd {t_bwd:8.1f}us | "
          f"sdpa fwd {t_sdpa_fwd:8.1f}us ({flops_fwd / t_sdpa_fwd / 1e6:6.1f} TF)  "
          f"fwd+bwd {t_sdpa_fb:8.1f}us")


if __name__ == "__main__":
    run_shape(8, 16, 2048, 64)    # bloom-560m
    run_shape(8, 32, 2048, 128)   # bloom-7b1
