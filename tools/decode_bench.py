"""Serving microbenchmark: KV-cached greedy decode throughput.

Usage (GPU box): python tools/decode_bench.py [--model bloom-560m]
Prints prefill tokens/s and per-step decode latency / tokens/s.
"""
import argparse
import os
import sys
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="bloom-560m",
                   choices=["bloom-560m", "bloom-1b7", "bloom-7b1", "bloom-tiny"])
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--prompt-len", type=int, default=512)
    p.add_argument("--new-tokens", type=int, default=64)
    p.add_argument("--graph", action="store_true",
                   help="also time the hipGraph-captured decode loop")
    args = p.parse_args()

    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "2977")

    from pipegoose_amd import ParallelContext
    from pipegoose_amd.models import bloom as M

    ctx = ParallelContext.from_torch()
    cfg = {"bloom-560m": M.bloom_560m, "bloom-1b7": M.bloom_1b7,
           "bloom-7b1": M.bloom_7b1, "bloom-tiny": M.bloom_tiny}[args.model]()
    use_gpu = torch.cuda.is_available()
    dev = "cuda" if use_gpu else "cpu"
    dt = torch.bfloat16 if use_gpu else torch.float32
    torch.manual_seed(0)
    model = M.BloomForCausalLM(cfg, ctx).to(dev, dt).eval()

    B, P, N = args.batch, args.prompt_len, args.new_tokens
    ids = torch.randint(0, cfg.vocab_size, (B, P), device=dev)

    with torch.no_grad():
        # warmup
        model.generate(ids[:, :64], max_new_tokens=4)
        if use_gpu:
            torch.cuda.synchronize()

        past = model.new_kv_cache(B, P + N)  # static: no per-step history cat
        t0 = time.perf_counter()
        logits, _ = model(ids, past=past, use_cache=True)
        if use_gpu:
            torch.cuda.synchronize()
        t_prefill = time.perf_counter() - t0

        nxt = logits[:, -1].argmax(-1, keepdim=True)
        t0 = time.perf_counter()
        for _ in range(N):
            logits, _ = model(nxt, past=past, use_cache=True)
            nxt = logits[:, -1].argmax(-1, keepdim=True)
        if use_gpu:
            torch.cuda.synchronize()
        t_decode = time.perf_counter() - t0

    print(f"{args.model} B{B} prompt{P} new{N}: "
          f"prefill {B * P / t_prefill:,.0f} tok/s ({t_prefill * 1e3:.1f} ms) | "
          f"decode {t_decode / N * 1e3:.2f} ms/step "
          f"= {B * N / t_decode:,.0f} tok/s")

    # The public entry point (generate() auto-routes greedy tp=1 CUDA decode
    # through the hipGraph decoder) — this is the number a user of
    # model.generate() actually sees.
    with torch.no_grad():
        model.generate(ids, max_new_tokens=N)  # warmup (+ graph capture)
        if use_gpu:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        model.generate(ids, max_new_tokens=N)
        if use_gpu:
            torch.cuda.synchronize()
        t_gen = time.perf_counter() - t0
    print(f"  generate() [default path]: includes prefill; "
          f"{(t_gen - t_prefill) / N * 1e3:.2f} ms/step "
          f"= {B * N / (t_gen - t_prefill):,.0f} tok/s")

    if args.graph:
        from pipegoose_amd.models.graph_decode import GraphDecoder
        dec = GraphDecoder(model, batch_size=B, max_len=P + 2 * N + 8)
        with torch.no_grad():
            dec.generate(ids, max_new_tokens=N)  # warmup + capture
            if use_gpu:
                torch.cuda.synchronize()
            t0 = time.perf_counter()
            out = dec.generate(ids, max_new_tokens=N)
            if use_gpu:
                torch.cuda.synchronize()
            t_graph = time.perf_counter() - t0
        mode = "hipGraph" if dec._graph is not None else "eager-fallback"
        print(f"  graph decode ({mode}): includes prefill; "
              f"{(t_graph - t_prefill) / N * 1e3:.2f} ms/step "
              f"= {B * N / (t_graph - t_prefill):,.0f} tok/s")
    ctx.destroy()


if __name__ == "__main__":
    main()
