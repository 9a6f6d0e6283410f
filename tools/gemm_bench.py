"""Hand-written MFMA GEMM vs hipBLASLt (torch.matmul) on the TP-linear shapes.

Usage (GPU box): python tools/gemm_bench.py [--markdown]
Prints TF/s for both paths per (M, N, K) of the bench models at tp1/tp2 —
this is the match-or-fallback evidence table; the
summary is committed under profiles/.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))
from pipegoose_amd.ops import get_extension  # noqa: E402

# (name, M, N, K): M = B*S of the bench configs (bloom-560m B8 S2048;
# bloom-7b1 B2 S2048); N/K from the four linear shapes per block at tp1/tp2.
SHAPES = [
    ("560m qkv tp1",      16384, 3072, 1024),
    ("560m dense tp1",    16384, 1024, 1024),
    ("560m h4h tp1",      16384, 4096, 1024),
    ("560m 4hh tp1",      16384, 1024, 4096),
    ("560m qkv tp2",      16384, 1536, 1024),
    ("560m h4h tp2",      16384, 2048, 1024),
    ("560m 4hh tp2",      16384, 1024, 2048),
    ("7b1 qkv tp1",        4096, 12288, 4096),
    ("7b1 dense tp1",      4096, 4096, 4096),
    ("7b1 h4h tp1",        4096, 16384, 4096),
    ("7b1 4hh tp1",        4096, 4096, 16384),
    ("7b1 h4h tp2",        4096, 8192, 4096),
    ("7b1 4hh tp2",        4096, 4096, 8192),
    ("560m lmhead tp2",   16384, 125440, 1024),
    ("square 4096", 4096, 4096, 4096),
    ("square 8192", 8192, 8192, 8192),
]


def bench(fn, iters=10, warmup=3):
    for _ in range(warmup):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--markdown", action="store_true")
    args = ap.parse_args()
    ext = get_extension(required=True)
    rows = []
    for name, M, N, K in SHAPES:
        torch.manual_seed(0)
        A = (torch.rand(M, K, device="cuda", dtype=torch.float32) * 2 - 1).bfloat16()
        B = (torch.rand(N, K, device="cuda", dtype=torch.float32) * 2 - 1).bfloat16()
        flops = 2.0 * M * N * K

        t_lib = bench(lambda: torch.matmul(A, B.t()))
        supported = M % 128 == 0 and N % 128 == 0 and K % 64 == 0
        if supported:
            C_ours = ext.gemm_bt(A, B)
            C_ref = torch.matmul(A, B.t())
            err = (C_ours.float() - C_ref.float()).abs().max().item()
            rel = err / max(C_ref.float().abs().max().item(), 1.0)
            t_ours = bench(lambda: ext.gemm_bt(A, B))
        else:
            t_ours, rel = float("nan"), float("nan")
        rows.append((name, M, N, K, flops / t_lib / 1e12,
                     flops / t_ours / 1e12 if supported else float("nan"),
                     rel))
        del A, B
        torch.cuda.empty_cache()

    hdr = f"{'shape':18s} {'M':>6s} {'N':>7s} {'K':>6s} {'hipBLASLt TF':>12s} {'ours TF':>9s} {'relerr':>8s} {'winner':>9s}"
    sep = "|---" * 8 + "|" if args.markdown else ""
    if args.markdown:
        print("| shape | M | N | K | hipBLASLt TF | ours TF | relerr | winner |")
        print(sep)
    else:
        print(hdr)
    for name, M, N, K, tf_lib, tf_ours, rel in rows:
        win = "ours" if tf_ours == tf_ours and tf_ours > tf_lib else "hipBLASLt"
        if args.markdown:
            print(f"| {name} | {M} | {N} | {K} | {tf_lib:.0f} | "
                  f"{tf_ours:.0f} | {rel:.1e} | {win} |")
        else:
            print(f"{name:18s} {M:6d} {N:7d} {K:6d} {tf_lib:12.0f} "
                  f"{tf_ours:9.0f} {rel:8.1e} {win:>9s}")


if __name__ == "__main__":
    main()
