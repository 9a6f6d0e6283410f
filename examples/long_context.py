"""Context-parallel long-context training on MI355X.

Sequences shard along S over the CONTEXT ring; attention runs as
rotation-based ring attention with bounded per-rank memory (each rank ever
holds two KV blocks), so the trainable sequence length scales with the
number of GPUs while activations stay at S/cp per rank.

Launch (e.g. 4 GPUs, one 32k-token sequence split 4 ways):
    torchrun --nproc-per-node 4 --master-addr 127.0.0.1 \
        examples/long_context.py --cp 4 --seq-len 32768

Works the same on CPU/gloo for a smoke test:
    torchrun --nproc-per-node 2 examples/long_context.py \
        --cp 2 --seq-len 256 --model tiny --steps 2
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pipegoose_amd import ParallelContext, ParallelMode
from pipegoose_amd.models.bloom import (BloomForCausalLM, bloom_560m,
                                        bloom_tiny)
from pipegoose_amd.nn import DataParallel
from pipegoose_amd.optim.fused_adamw import FusedAdamW


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--cp", type=int, default=2)
    ap.add_argument("--seq-len", type=int, default=8192,
                    help="FULL sequence length (sharded S/cp per rank)")
    ap.add_argument("--model", default="560m", choices=["560m", "tiny"])
    ap.add_argument("--steps", type=int, default=5)
    ap.add_argument("--batch", type=int, default=1)
    args = ap.parse_args()
    assert args.seq_len % args.cp == 0

    ctx = ParallelContext.from_torch(context_parallel_size=args.cp)
    cfg = bloom_560m() if args.model == "560m" else bloom_tiny()
    cfg.context_parallel = True

    use_gpu = torch.cuda.is_available()
    dtype = torch.bfloat16 if use_gpu else torch.float32
    torch.manual_seed(7)
    model = BloomForCausalLM(cfg, ctx).to(device=ctx.device, dtype=dtype)
    # parameters replicate over the CONTEXT ring -> grads all-reduce like DP
    DataParallel(model, ctx, mode=ParallelMode.CONTEXT).parallelize()
    optim = FusedAdamW(model.parameters(), lr=1e-4) if use_gpu \
        else torch.optim.AdamW(model.parameters(), lr=1e-4)

    cp_rank = ctx.get_local_rank(ParallelMode.CONTEXT)
    S_local = args.seq_len // args.cp
    g = torch.Generator().manual_seed(1234)  # same full sequence every rank
    for step in range(args.steps):
        full = torch.randint(0, cfg.vocab_size, (args.batch, args.seq_len),
                             generator=g)
        ids = full[:, cp_rank * S_local:(cp_rank + 1) * S_local] \
            .to(ctx.device)
        optim.zero_grad(set_to_none=True)
        loss = model(ids, labels=ids)
        loss.backward()
        optim.step()
        if ctx.get_global_rank() == 0:
            print(f"step {step}: shard loss {loss.item():.4f} "
                  f"(S={args.seq_len} over cp={args.cp})", flush=True)
    ctx.destroy()


if __name__ == "__main__":
    main()
