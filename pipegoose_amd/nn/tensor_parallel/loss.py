"""Vocab-parallel cross-entropy (reference: nn/tensor_parallel/loss.py).

Streaming CE over a vocab-sharded logit tensor [B, S, V/tp]:
  1. local row max            -> all-reduce MAX over TENSOR
  2. local target-logit pick  -> all-reduce SUM  (zero where target off-shard)
  3. local sum-exp            -> all-reduce SUM
  loss = log(sum_exp) - (target_logit - max)

Backward writes (softmax - one_hot) * grad / N in place, Megatron-style.
On GPU steps 1-3 fuse into one HIP kernel pass over the shard
(pipegoose_amd.ops.cross_entropy) followed by two RCCL all-reduces.
"""
import torch
from torch import nn
from torch.distributed import ReduceOp

from pipegoose_amd.distributed import functional as F
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.tensor_parallel._utils import VocabUtility


class _VocabParallelCrossEntropy(torch.autograd.Function):
    @staticmethod
    def forward(ctx, logits: torch.Tensor, targets: torch.Tensor,
                parallel_context: ParallelContext):
        # logits: [N, V_local] fp32/bf16; targets: [N] int64 (global vocab ids)
        rank = parallel_context.get_local_rank(ParallelMode.TENSOR)
        partition = logits.size(-1)
        vocab_start, vocab_end = VocabUtility.get_vocab_range_from_partition_size(
            partition, rank)

        logits_max = logits.max(dim=-1).values
        F.all_reduce(logits_max, op=ReduceOp.MAX, parallel_context=parallel_context,
                     parallel_mode=ParallelMode.TENSOR)

        shifted = logits.float() - logits_max.unsqueeze(-1)

        mask = (targets < vocab_start) | (targets >= vocab_end)
        local_targets = (targets - vocab_start).masked_fill(mask, 0)
        target_logits = shifted.gather(-1, local_targets.unsqueeze(-1)).squeeze(-1)
        target_logits = target_logits.masked_fill(mask, 0.0)
        F.all_reduce(target_logits, parallel_context=parallel_context,
                     parallel_mode=ParallelMode.TENSOR)

        exp_logits = shifted.exp()
        sum_exp = exp_logits.sum(dim=-1)
        F.all_reduce(sum_exp, parallel_context=parallel_context,
                     parallel_mode=ParallelMode.TENSOR)

        loss = torch.log(sum_exp) - target_logits

        softmax = exp_logits / sum_exp.unsqueeze(-1)
        ctx.save_for_backward(softmax, mask, local_targets)
        ctx.logits_dtype = logits.dtype
        return loss

    @staticmethod
    def backward(ctx, grad_output):
        softmax, mask, local_targets = ctx.saved_tensors
        grad = softmax  # [N, V_local] fp32; modified in place (it is ours)
        one_hot = (~mask).float()
        grad.scatter_add_(-1, local_targets.unsqueeze(-1), -one_hot.unsqueeze(-1))
        grad.mul_(grad_output.unsqueeze(-1))
        return grad.to(ctx.logits_dtype), None, None


class VocabParallelCrossEntropy(nn.Module):
    def __init__(self, parallel_context: ParallelContext, reduction: str = "mean"):
        super().__init__()
        self.parallel_context = parallel_context
        self.reduction = reduction

    def forward(self, logits: torch.Tensor, targets: torch.Tensor) -> torch.Tensor:
        # accept [B, S, V_local] + [B, S] or flat [N, V_local] + [N]
        orig_shape = targets.shape
        logits = logits.reshape(-1, logits.size(-1))
        targets = targets.reshape(-1)
        if logits.is_cuda:
            # fused single-pass HIP kernel + 2 allreduces (ops/cross_entropy)
            from pipegoose_amd.distributed.parallel_mode import ParallelMode
            from pipegoose_amd.nn.tensor_parallel._utils import VocabUtility
            from pipegoose_amd.ops.cross_entropy import fused_cross_entropy
            rank = self.parallel_context.get_local_rank(ParallelMode.TENSOR)
            start, end = VocabUtility.get_vocab_range_from_partition_size(
                logits.size(-1), rank)
            loss = fused_cross_entropy(logits, targets, start, end,
                                       self.parallel_context, reduction="none")
        else:
            loss = _VocabParallelCrossEntropy.apply(logits, targets,
                                                    self.parallel_context)
        valid = targets >= 0  # ignore_index support (negative labels)
        if not bool(valid.all()):
            loss = loss * valid
            if self.reduction == "mean":
                return loss.sum() / valid.sum().clamp(min=1)
        if self.reduction == "mean":
            return loss.mean()
        if self.reduction == "sum":
            return loss.sum()
        return loss.reshape(orig_shape)
