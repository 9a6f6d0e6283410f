"""KV-cached batch decode (serving path).

Launch (TP2):
    torchrun --nproc-per-node 2 --master-addr 127.0.0.1 examples/serving.py
CPU demo:
    python examples/serving.py --tiny
"""
import argparse

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import time

import torch

from pipegoose_amd import ParallelContext
from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_560m, bloom_tiny


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--batch", type=int, default=8)
    p.add_argument("--prompt-len", type=int, default=256)
    p.add_argument("--new-tokens", type=int, default=32)
    p.add_argument("--tiny", action="store_true")
    args = p.parse_args()

    ctx = ParallelContext.from_torch(
        tensor_parallel_size=int(__import__("os").environ.get("WORLD_SIZE", 1)))
    cfg = bloom_tiny() if args.tiny else bloom_560m()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dt = torch.bfloat16 if dev == "cuda" else torch.float32
    torch.manual_seed(0)
    model = BloomForCausalLM(cfg, ctx).to(dev, dt).eval()

    ids = torch.randint(0, cfg.vocab_size, (args.batch, args.prompt_len),
                        device=dev)
    t0 = time.perf_counter()
    out = model.generate(ids, max_new_tokens=args.new_tokens)  # static KV cache
    if dev == "cuda":
        torch.cuda.synchronize()
    dt_s = time.perf_counter() - t0
    if ctx.get_global_rank() == 0:
        n_new = out.size(1) - ids.size(1)
        print(f"generated {n_new} tokens x batch {args.batch} "
              f"in {dt_s * 1e3:.1f} ms "
              f"({args.batch * n_new / dt_s:,.0f} tokens/s decode incl. prefill)")

    if ctx.get_world_size() == 1:
        # single-GPU fast path: hipGraph-captured decode loop (2.24x measured
        # on MI355X — models/graph_decode.py)
        from pipegoose_amd.models.graph_decode import GraphDecoder
        dec = GraphDecoder(model, batch_size=args.batch,
                           max_len=args.prompt_len + 2 * args.new_tokens + 8)
        dec.generate(ids, max_new_tokens=args.new_tokens)  # warmup + capture
        if dev == "cuda":
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        out2 = dec.generate(ids, max_new_tokens=args.new_tokens)
        if dev == "cuda":
            torch.cuda.synchronize()
        dt_g = time.perf_counter() - t0
        mode = "hipGraph" if dec._graph is not None else "eager"
        print(f"graph decoder ({mode}): {args.batch * out2.size(1) / dt_g:,.0f} "
              f"tokens/s incl. prefill")
    ctx.destroy()


if __name__ == "__main__":
    main()
