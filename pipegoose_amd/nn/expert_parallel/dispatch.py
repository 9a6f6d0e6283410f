"""Token all-to-all dispatch for expert parallelism.

The reference had no all-to-all: its Experts layer ran every rank over the
full token set with boolean masks and combined with an all-reduce
(reference nn/expert_parallel/experts.py:75-102) — O(world) redundant FLOPs
and an [N, H] all-reduce.  Here tokens physically move: each token is sent
once to the single rank that owns its expert and its result is sent back, so
the wire traffic is 2×[N, H] split across the xGMI point-to-point links and
each expert runs only its own tokens.  On RCCL the exchange is one fused
``all_to_all_single``; backward is the reverse exchange of gradients.
"""
from typing import List

import torch

from pipegoose_amd.distributed import functional as F
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class _AllToAllVariable(torch.autograd.Function):
    """Differentiable variable-split all-to-all along dim 0.  The gradient of
    "send my split i to rank i" is "receive grad split i from rank i" — the
    same exchange with in/out splits swapped."""

    @staticmethod
    def forward(ctx, tensor, in_splits: List[int], out_splits: List[int],
                parallel_context, parallel_mode):
        ctx.in_splits = in_splits
        ctx.out_splits = out_splits
        ctx.parallel_context = parallel_context
        ctx.parallel_mode = parallel_mode
        out, _ = F.all_to_all_variable(
            tensor, in_splits, parallel_context=parallel_context,
            parallel_mode=parallel_mode, out_splits=out_splits)
        return out

    @staticmethod
    def backward(ctx, grad):
        gin, _ = F.all_to_all_variable(
            grad.contiguous(), ctx.out_splits, parallel_context=ctx.parallel_context,
            parallel_mode=ctx.parallel_mode, out_splits=ctx.in_splits)
        return gin, None, None, None, None


class AllToAllDispatcher:
    """Route tokens to expert-owner ranks and back over one parallel group.

    Experts are sharded contiguously: rank r owns global experts
    [r*local, (r+1)*local).  ``dispatch`` returns this rank's received tokens
    sorted by LOCAL expert id plus everything needed to ``combine`` back into
    the original token order.
    """

    def __init__(self, num_experts: int, parallel_context: ParallelContext,
                 parallel_mode: ParallelMode = ParallelMode.TENSOR):
        self.num_experts = num_experts
        self.ctx = parallel_context
        self.mode = parallel_mode
        self.world_size = parallel_context.get_world_size(parallel_mode)
        assert num_experts % self.world_size == 0
        self.num_local = num_experts // self.world_size

    def dispatch(self, tokens: torch.Tensor, expert_idx: torch.Tensor):
        """tokens [N, H]; expert_idx [N] global expert id (top-1).

        Returns (recv_tokens [M, H] sorted by local expert, recv_local_idx [M],
        state) where state carries the permutation + splits for combine().
        """
        assert tokens.dim() == 2 and expert_idx.numel() == tokens.size(0)
        expert_idx = expert_idx.reshape(-1)
        # sort by destination: global expert id order == (dest rank, local id)
        send_order = torch.argsort(expert_idx, stable=True)
        send_counts = torch.bincount(expert_idx, minlength=self.num_experts)
        in_splits = send_counts.reshape(self.world_size, self.num_local) \
                               .sum(dim=1).tolist()
        sorted_tokens = tokens[send_order]
        sorted_experts = expert_idx[send_order]

        # exchange split sizes once (int64 all-gather), reuse for both
        # directions and for the id exchange
        out_splits = F.exchange_splits(in_splits, self.ctx, self.mode)

        recv_tokens = _AllToAllVariable.apply(
            sorted_tokens, in_splits, out_splits, self.ctx, self.mode)
        with torch.no_grad():
            recv_global_idx, _ = F.all_to_all_variable(
                sorted_experts, in_splits, parallel_context=self.ctx,
                parallel_mode=self.mode, out_splits=out_splits)
        rank = self.ctx.get_local_rank(self.mode)
        recv_local_idx = recv_global_idx - rank * self.num_local

        # group received tokens by local expert for contiguous expert batches
        local_order = torch.argsort(recv_local_idx, stable=True)
        state = {
            "send_order": send_order,
            "in_splits": in_splits,
            "out_splits": out_splits,
            "local_order": local_order,
            "n_tokens": tokens.size(0),
        }
        return recv_tokens[local_order], recv_local_idx[local_order], state

    def combine(self, expert_out: torch.Tensor, state) -> torch.Tensor:
        """Inverse of dispatch: un-group, exchange back, un-sort."""
        inv_local = torch.empty_like(state["local_order"])
        inv_local[state["local_order"]] = torch.arange(
            expert_out.size(0), device=expert_out.device)
        unsorted = expert_out[inv_local]
        back = _AllToAllVariable.apply(
            unsorted, state["out_splits"], state["in_splits"], self.ctx, self.mode)
        inv_send = torch.empty_like(state["send_order"])
        inv_send[state["send_order"]] = torch.arange(
            state["n_tokens"], device=expert_out.device)
        return back[inv_send]
