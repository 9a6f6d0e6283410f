"""Serving-stack throughput: concurrent requests through BatchingEngine.

Usage (GPU box): python tools/serve_bench.py [--model bloom-560m]
Fires N concurrent same-shape requests at the engine and reports
end-to-end request and generated-token throughput (the engine coalesces
them into batched hipGraph decodes).
"""
import argparse
import os
import sys
import threading
import time

import torch

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--model", default="bloom-560m",
                   choices=["bloom-560m", "bloom-1b7", "bloom-tiny"])
    p.add_argument("--requests", type=int, default=64)
    p.add_argument("--prompt-len", type=int, default=128)
    p.add_argument("--new-tokens", type=int, default=32)
    p.add_argument("--max-batch", type=int, default=8)
    args = p.parse_args()

    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "2978")

    from pipegoose_amd import ParallelContext
    from pipegoose_amd.models import bloom as M
    from pipegoose_amd.serve import BatchingEngine

    ctx = ParallelContext.from_torch()
    cfg = {"bloom-560m": M.bloom_560m, "bloom-1b7": M.bloom_1b7,
           "bloom-tiny": M.bloom_tiny}[args.model]()
    use_gpu = torch.cuda.is_available()
    dev = "cuda" if use_gpu else "cpu"
    dt = torch.bfloat16 if use_gpu else torch.float32
    torch.manual_seed(0)
    model = M.BloomForCausalLM(cfg, ctx).to(dev, dt).eval()
    engine = BatchingEngine(model, parallel_context=ctx,
                            max_batch=args.max_batch, max_wait_ms=20.0)

    prompts = [torch.randint(0, cfg.vocab_size, (args.prompt_len,))
               for _ in range(args.requests)]
    engine.submit(prompts[0], max_new_tokens=args.new_tokens)  # warm + capture

    def fire(ids):
        engine.submit(ids, max_new_tokens=args.new_tokens)

    t0 = time.perf_counter()
    threads = [threading.Thread(target=fire, args=(q,)) for q in prompts]
    for t in threads:
        t.start()
    for t in threads:
        t.join()
    dt_s = time.perf_counter() - t0

    toks = args.requests * args.new_tokens
    print(f"{args.model} {args.requests} reqs P{args.prompt_len} "
          f"new{args.new_tokens} max_batch{args.max_batch}: "
          f"{dt_s:.2f}s = {args.requests / dt_s:.1f} req/s, "
          f"{toks / dt_s:,.0f} gen tok/s "
          f"({engine.n_batches} batches for {engine.n_requests} requests)")
    engine.shutdown()
    ctx.destroy()


if __name__ == "__main__":
    main()
