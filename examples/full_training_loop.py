"""The full user journey in one file: data → parallelize → Trainer with
checkpointing, logging, throughput metering and a hang watchdog.

Launch (TP2×DP2 on 4 GPUs):
    torchrun --nproc-per-node 4 --master-addr 127.0.0.1 \
        examples/full_training_loop.py --tp 2
CPU demo:
    python examples/full_training_loop.py --tiny --steps 5
"""
import argparse
import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from pipegoose_amd import ParallelContext
from pipegoose_amd.data import SyntheticLMDataset, build_dataloader
from pipegoose_amd.models.bloom import (BloomForCausalLM, bloom_560m,
                                        bloom_tiny, make_causal_lm_loss)
from pipegoose_amd.nn import DataParallel
from pipegoose_amd.optim import DistributedOptimizer
from pipegoose_amd.trainer import Trainer
from pipegoose_amd.trainer.callback import CheckpointCallback


def main():
    p = argparse.ArgumentParser()
    p.add_argument("--tp", type=int, default=1)
    p.add_argument("--steps", type=int, default=20)
    p.add_argument("--micro-batch", type=int, default=4)
    p.add_argument("--seq-len", type=int, default=128)
    p.add_argument("--ckpt-dir", default="/tmp/pipegoose_ckpt")
    p.add_argument("--tiny", action="store_true")
    args = p.parse_args()

    world = int(os.environ.get("WORLD_SIZE", 1))
    ctx = ParallelContext.from_torch(
        tensor_parallel_size=args.tp,
        data_parallel_size=max(1, world // args.tp))
    cfg = bloom_tiny() if args.tiny else bloom_560m()
    dev = "cuda" if torch.cuda.is_available() else "cpu"
    dt = torch.bfloat16 if dev == "cuda" else torch.float32

    torch.manual_seed(0)  # identical init on every rank (TP-aware model)
    model = BloomForCausalLM(cfg, ctx).to(dev, dt)
    model = DataParallel(model, ctx).parallelize()

    dataset = SyntheticLMDataset(n_samples=1024, seq_len=args.seq_len,
                                 vocab_size=cfg.vocab_size)
    loader = build_dataloader(dataset, args.micro_batch,
                              parallel_context=ctx)

    loss_fn = make_causal_lm_loss(ctx)
    optim = DistributedOptimizer(
        torch.optim.AdamW(model.parameters(), lr=2e-4), ctx,
        grad_reduce="shard")

    def to_dev(batch):
        return {k: v.to(dev) for k, v in batch.items()}

    trainer = Trainer(
        model, optim,
        loss_fn=lambda logits, labels: loss_fn(logits, labels),
        parallel_context=ctx,
        max_grad_norm=1.0,
        log_interval=5,
        hang_timeout_s=600,
        callbacks=[CheckpointCallback(args.ckpt_dir, every_steps=50)])
    trainer.fit(map(to_dev, iter(loader)), max_steps=args.steps)
    if ctx.get_global_rank() == 0:
        print(f"done: step {trainer.state.global_step}, "
              f"last loss {trainer.state.last_loss:.4f}, "
              f"checkpoints in {args.ckpt_dir}")
    ctx.destroy()


if __name__ == "__main__":
    main()
