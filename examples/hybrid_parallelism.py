"""Hybrid 3D-parallel training on MI355X — the reference's headline example
(reference examples/hybrid_parallelism.py), MI355X-native.

Launch (single 8-GPU node):
    torchrun --nproc-per-node 8 --master-addr 127.0.0.1 \
        examples/hybrid_parallelism.py --tp 2 --pp 2 --dp 2
"""
import argparse

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))


import torch

from pipegoose_amd import ParallelContext
from pipegoose_amd.models.bloom import (BloomForCausalLM, bloom_560m,
                                        bloom_tiny, make_causal_lm_loss)
from pipegoose_amd.nn import DataParallel
from pipegoose_amd.nn.pipeline_parallel import PipelineParallel
from pipegoose_amd.optim import DistributedOptimizer
from pipegoose_amd.trainer import Trainer


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tp", type=int, default=2)
    p.add_argument("--pp", type=int, default=1)
    p.add_argument("--dp", type=int, default=2)
    p.add_argument("--steps", type=int, default=50)
    p.add_argument("--tiny", action="store_true", help="tiny config (CPU demo)")
    args = p.parse_args()

    ctx = ParallelContext.from_torch(
        tensor_parallel_size=args.tp,
        pipeline_parallel_size=args.pp,
        data_parallel_size=args.dp,
    )
    cfg = bloom_tiny() if args.tiny else bloom_560m()
    model = BloomForCausalLM(cfg, ctx)  # TP-sharded by construction
    if args.pp > 1:
        model = PipelineParallel(
            model, ctx, n_microbatches=4,
            loss_fn=make_causal_lm_loss(ctx)).parallelize()
    if torch.cuda.is_available():
        model = model.to("cuda", torch.bfloat16)
    model = DataParallel(model, ctx).parallelize()

    optim = DistributedOptimizer(
        torch.optim.AdamW(model.parameters(), lr=1e-4), ctx)

    trainer = Trainer(model, optim, parallel_context=ctx, max_grad_norm=1.0)

    def batches():
        g = torch.Generator().manual_seed(ctx.get_global_rank())
        dev = "cuda" if torch.cuda.is_available() else "cpu"
        while True:
            ids = torch.randint(0, cfg.vocab_size, (4, 512), generator=g).to(dev)
            yield {"input_ids": ids, "labels": ids}

    trainer.fit(batches(), max_steps=args.steps)
    ctx.destroy()


if __name__ == "__main__":
    main()
