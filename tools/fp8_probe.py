"""Probe fp8 GEMM support on this ROCm/PyTorch build (gfx950).

Checks torch._scaled_mm with e4m3/e5m2 operands (tensor-wise scales) and
times it against bf16 torch.mm on the TP-linear bench shapes.

Usage (GPU box): python tools/fp8_probe.py
"""
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def bench(fn, iters=30, warmup=8):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    if not torch.cuda.is_available():
        print("no GPU")
        return
    print("fp8 dtypes:", hasattr(torch, "float8_e4m3fn"),
          hasattr(torch, "float8_e5m2"))
    dev = "cuda"
    shapes = [(16384, 1024, 4096), (16384, 4096, 1024),
              (16384, 4096, 14336), (16384, 14336, 4096)]
    for (m, k, n) in shapes:
        a = torch.randn(m, k, device=dev, dtype=torch.bfloat16)
        b = torch.randn(n, k, device=dev, dtype=torch.bfloat16)
        t_bf = bench(lambda: a @ b.t())
        tf_bf = 2 * m * n * k / t_bf / 1e12
        try:
            a8 = a.to(torch.float8_e4m3fn)
            b8 = b.to(torch.float8_e4m3fn)
            sa = torch.tensor(1.0, device=dev)
            sb = torch.tensor(1.0, device=dev)
            out = torch._scaled_mm(a8, b8.t(), scale_a=sa, scale_b=sb,
                                   out_dtype=torch.bfloat16)
            t_f8 = bench(lambda: torch._scaled_mm(
                a8, b8.t(), scale_a=sa, scale_b=sb,
                out_dtype=torch.bfloat16))
            tf_f8 = 2 * m * n * k / t_f8 / 1e12
            ref = (a.float() @ b.float().t())
            rel = (out.float() - ref).abs().mean() / ref.abs().mean()
            print(f"M{m} K{k} N{n}: bf16 {tf_bf:7.1f} TF | "
                  f"fp8 {tf_f8:7.1f} TF ({tf_f8 / tf_bf:.2f}x) "
                  f"relerr {rel.item():.3e}")
        except Exception as e:
            print(f"M{m} K{k} N{n}: bf16 {tf_bf:7.1f} TF | "
                  f"fp8 FAILED: {type(e).__name__}: {e}")


if __name__ == "__main__":
    main()
