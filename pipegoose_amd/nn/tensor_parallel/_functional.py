"""Autograd-wrapped TP collectives (reference: nn/tensor_parallel/_functional.py).

Four conjugate pairs over the TENSOR group:
  _Broadcast: identity fwd / all-reduce bwd   (column-linear input)
  _Gather:    all-gather fwd / chunk bwd      (gather_output)
  _Scatter:   chunk fwd / all-gather bwd      (row-linear input)
  _Reduce:    all-reduce fwd / identity bwd   (row-linear output)
"""
import torch

from pipegoose_amd.distributed import functional as F
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class _Broadcast(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, parallel_context):
        ctx.parallel_context = parallel_context
        return tensor

    @staticmethod
    def backward(ctx, grad):
        grad = grad.contiguous()
        F.all_reduce(grad, parallel_context=ctx.parallel_context,
                     parallel_mode=ParallelMode.TENSOR)
        return grad, None


class _Gather(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, dim, parallel_context):
        ctx.dim = dim
        ctx.parallel_context = parallel_context
        return F.all_gather(tensor, dim=dim, parallel_context=parallel_context,
                            parallel_mode=ParallelMode.TENSOR)

    @staticmethod
    def backward(ctx, grad):
        pc = ctx.parallel_context
        rank = pc.get_local_rank(ParallelMode.TENSOR)
        world = pc.get_world_size(ParallelMode.TENSOR)
        if world == 1:
            return grad, None, None
        return grad.chunk(world, dim=ctx.dim)[rank].contiguous(), None, None


class _Scatter(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, dim, parallel_context):
        ctx.dim = dim
        ctx.parallel_context = parallel_context
        return F.scatter(tensor, dim=dim, parallel_context=parallel_context,
                         parallel_mode=ParallelMode.TENSOR).contiguous()

    @staticmethod
    def backward(ctx, grad):
        out = F.all_gather(grad, dim=ctx.dim, parallel_context=ctx.parallel_context,
                           parallel_mode=ParallelMode.TENSOR)
        return out, None, None


class _Reduce(torch.autograd.Function):
    @staticmethod
    def forward(ctx, tensor, parallel_context):
        tensor = tensor.contiguous()
        F.all_reduce(tensor, parallel_context=parallel_context,
                     parallel_mode=ParallelMode.TENSOR)
        return tensor

    @staticmethod
    def backward(ctx, grad):
        return grad, None


class _AllGatherSP(torch.autograd.Function):
    """Sequence-parallel boundary entering a TP region: forward all-gathers
    the sequence shards ([B, S/tp, H] → [B, S, H]); backward REDUCE-SCATTERS
    the gradient — this replaces _Broadcast's backward all-reduce, so a
    column-linear in SP mode must use this instead of broadcast (Megatron-SP;
    absent in the reference, SURVEY.md §5 'Long-context / sequence
    parallelism')."""

    @staticmethod
    def forward(ctx, tensor, dim, parallel_context):
        ctx.dim = dim
        ctx.parallel_context = parallel_context
        return F.all_gather(tensor.contiguous(), dim=dim,
                            parallel_context=parallel_context,
                            parallel_mode=ParallelMode.TENSOR)

    @staticmethod
    def backward(ctx, grad):
        out = F.reduce_scatter(grad.contiguous(), dim=ctx.dim,
                               parallel_context=ctx.parallel_context,
                               parallel_mode=ParallelMode.TENSOR)
        return out, None, None


class _ReduceScatterSP(torch.autograd.Function):
    """Sequence-parallel boundary leaving a TP region: forward reduce-scatters
    the partial sums along the sequence dim ([B, S, H] partial → [B, S/tp, H]
    complete); backward all-gathers the gradient shards.  Replaces _Reduce's
    forward all-reduce — same bytes on the xGMI wire, but each rank keeps only
    its sequence shard (activation memory ÷ tp)."""

    @staticmethod
    def forward(ctx, tensor, dim, parallel_context):
        ctx.dim = dim
        ctx.parallel_context = parallel_context
        return F.reduce_scatter(tensor.contiguous(), dim=dim,
                                parallel_context=parallel_context,
                                parallel_mode=ParallelMode.TENSOR)

    @staticmethod
    def backward(ctx, grad):
        out = F.all_gather(grad.contiguous(), dim=ctx.dim,
                           parallel_context=ctx.parallel_context,
                           parallel_mode=ParallelMode.TENSOR)
        return out, None, None


def broadcast_to_tensor_group(tensor, parallel_context: ParallelContext):
    return _Broadcast.apply(tensor, parallel_context)


def gather_to_tensor_group(tensor, dim: int, parallel_context: ParallelContext):
    return _Gather.apply(tensor, dim, parallel_context)


def scatter_to_tensor_group(tensor, dim: int, parallel_context: ParallelContext):
    return _Scatter.apply(tensor, dim, parallel_context)


def reduce_to_tensor_group(tensor, parallel_context: ParallelContext):
    return _Reduce.apply(tensor, parallel_context)


def all_gather_sequence(tensor, parallel_context: ParallelContext, dim: int = 1):
    return _AllGatherSP.apply(tensor, dim, parallel_context)


def reduce_scatter_sequence(tensor, parallel_context: ParallelContext, dim: int = 1):
    return _ReduceScatterSP.apply(tensor, dim, parallel_context)
