"""Full fp8 linear (fwd+bwd, incl. quantization overhead) vs bf16.

The probe (fp8_probe.py) times the bare GEMM; this times what training
actually pays: dynamic quantization + 3 fp8 GEMMs + layout copies per
linear, against bf16 F.linear fwd+bwd.

Usage (GPU box): python tools/fp8_linear_bench.py
"""
import os
import sys
import time

import torch
import torch.nn as nn

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipegoose_amd.ops.fp8 import Fp8Linear  # noqa: E402


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
    shapes = [(16384, 1024, 4096), (16384, 4096, 1024),
              (16384, 4096, 14336), (16384, 14336, 4096)]
    for (m, k, n) in shapes:
        lin = nn.Linear(k, n).to("cuda", torch.bfloat16)
        x = torch.randn(m, k, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
        g = torch.randn(m, n, device="cuda", dtype=torch.bfloat16)

        def step():
            x.grad = None
            lin.weight.grad = None
            y = lin(x)
            y.backward(g)

        t_bf = bench(step)
        lin.__class__ = Fp8Linear
        t_f8 = bench(step)
        flops = 6 * m * n * k
        print(f"M{m} K{k} N{n}: bf16 fwd+bwd {t_bf * 1e6:8.1f}us "
              f"({flops / t_bf / 1e12:7.1f} TF) | fp8 {t_f8 * 1e6:8.1f}us "
              f"({flops / t_f8 / 1e12:7.1f} TF) -> {t_bf / t_f8:.2f}x")


if __name__ == "__main__":
    main()
