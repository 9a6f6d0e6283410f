"""Fused (vocab-parallel) cross-entropy op.

GPU: one online-softmax HIP pass per row over the local vocab shard, then
(for tp>1) MAX/SUM all-reduces of the tiny [N] stat vectors — same collective
structure as the reference's streaming CE (nn/tensor_parallel/loss.py) with
the local work fused into one kernel and no fp32 logit materialization.
CPU: eager reference implementation (the numerics oracle).
"""
from typing import Optional

import torch
from torch.distributed import ReduceOp

from pipegoose_amd.ops import get_extension


class _FusedCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits, targets, vocab_start, vocab_end, parallel_context):
        from pipegoose_amd.distributed import functional as F
        from pipegoose_amd.distributed.parallel_mode import ParallelMode
        ext = get_extension(required=True)
        row_max, row_sumexp, target_logit = ext.cross_entropy_fwd(
            logits, targets, vocab_start, vocab_end)
        if parallel_context is not None and \
                parallel_context.get_world_size(ParallelMode.TENSOR) > 1:
            local_max = row_max.clone()
            F.all_reduce(row_max, op=ReduceOp.MAX, parallel_context=parallel_context,
                         parallel_mode=ParallelMode.TENSOR)
            row_sumexp.mul_(torch.exp(local_max - row_max))
            # pack (sumexp, target_logit) into one all-reduce
            packed = torch.stack([row_sumexp, target_logit])
            F.all_reduce(packed, parallel_context=parallel_context,
                         parallel_mode=ParallelMode.TENSOR)
            row_sumexp, target_logit = packed[0], packed[1]
        loss = torch.log(row_sumexp) - (target_logit - row_max)
        ctx.save_for_backward(logits, targets, row_max, row_sumexp)
        ctx.vocab_range = (vocab_start, vocab_end)
        return loss

    @staticmethod
    def backward(ctx, grad_out):
        ext = get_extension(required=True)
        logits, targets, row_max, row_sumexp = ctx.saved_tensors
        vocab_start, vocab_end = ctx.vocab_range
        grad = ext.cross_entropy_bwd(
            logits, targets, row_max, row_sumexp,
            grad_out.contiguous().float(), vocab_start, vocab_end)
        return grad, None, None, None, None


def fused_cross_entropy(
    logits: torch.Tensor,
    targets: torch.Tensor,
    vocab_start: int = 0,
    vocab_end: Optional[int] = None,
    parallel_context=None,
    reduction: str = "mean",
) -> torch.Tensor:
    """logits [N, V_local] (bf16/fp32, cuda), targets [N] global vocab ids."""
    if vocab_end is None:
        vocab_end = vocab_start + logits.size(-1)
    loss = _FusedCrossEntropy.apply(logits.contiguous(), targets.contiguous(),
                                    vocab_start, vocab_end, parallel_context)
    # ignore_index (-100 style): negative targets contribute 0 loss and 0
    # grad (the masking is part of the autograd graph, so the backward kernel
    # sees gscale 0 on ignored rows).  Lets callers run CE over UNSLICED
    # logits with a padded shifted-label tensor instead of materializing
    # logits[:, :-1].contiguous() (8.2 GB per step at bloom vocab).
    valid = targets.reshape(-1) >= 0
    if not bool(valid.all()):
        loss = loss * valid
        if reduction == "mean":
            return loss.sum() / valid.sum().clamp(min=1)
    if reduction == "mean":
        return loss.mean()
    if reduction == "sum":
        return loss.sum()
    return loss
