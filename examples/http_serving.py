"""Serve a model over HTTP with dynamic micro-batching.

    python examples/serving.py --port 8000          # 1 process
    torchrun --nproc-per-node 2 --master-addr 127.0.0.1 \
        examples/serving.py --tp 2                  # TP serving

Then:

    curl -s localhost:8000/healthz
    curl -s -X POST localhost:8000/generate \
      -H 'content-type: application/json' \
      -d '{"input_ids": [[1,2,3,4]], "max_new_tokens": 16}'

Rank 0 owns the endpoint; under TP the other ranks join each batched
``generate()`` collectively (pipegoose_amd/serve.py).  On GPU the greedy
decode runs through the cached hipGraph decoder automatically.
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pipegoose_amd import ParallelContext
from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_560m, bloom_tiny
from pipegoose_amd.serve import serve


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--model", default="auto", choices=["auto", "560m", "tiny"])
    ap.add_argument("--tp", type=int, default=1)
    ap.add_argument("--host", default="127.0.0.1")
    ap.add_argument("--port", type=int, default=8000)
    ap.add_argument("--max-batch", type=int, default=8)
    args = ap.parse_args()

    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29591")
    ctx = ParallelContext.from_torch(tensor_parallel_size=args.tp)

    use_gpu = torch.cuda.is_available()
    if args.model == "auto":
        args.model = "560m" if use_gpu else "tiny"
    cfg = bloom_560m() if args.model == "560m" else bloom_tiny()
    torch.manual_seed(0)
    model = BloomForCausalLM(cfg, ctx).to(
        ctx.device, torch.bfloat16 if use_gpu else torch.float32).eval()
    # real deployments: nn.utils.load_full_state(model, <HF checkpoint>, ctx)

    serve(model, host=args.host, port=args.port, parallel_context=ctx,
          max_batch=args.max_batch)


if __name__ == "__main__":
    main()
