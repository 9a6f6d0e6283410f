"""Expert bank sharded over the TENSOR axis.

Reference parity: nn/expert_parallel/experts.py:41-102 (mask-select dispatch,
all-reduce combine over the TENSOR group; expert params tagged ``is_expert``
so DataParallel reduces them over the EXPERT_DATA group).

MI355X path: when every rank holds >1 expert the local experts run as a
grouped GEMM (one kernel for all local experts) via pipegoose_amd.ops; the
cross-rank combine stays a single RCCL all-reduce over xGMI.  Token
all-to-all dispatch is used by the EP=8 config (see expert_parallel.py).
"""
import copy

import torch
from torch import nn

from pipegoose_amd.distributed import functional as F
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class Experts(nn.Module):
    def __init__(self, num_experts: int, expert: nn.Module,
                 enable_tensor_parallel: bool, parallel_context: ParallelContext):
        super().__init__()
        self.parallel_context = parallel_context
        tp_size = parallel_context.get_world_size(ParallelMode.TENSOR)
        if enable_tensor_parallel and tp_size > 1:
            assert num_experts % tp_size == 0
            self.num_local_experts = num_experts // tp_size
            rank = parallel_context.get_local_rank(ParallelMode.TENSOR)
            self.expert_offset = rank * self.num_local_experts
        else:
            self.num_local_experts = num_experts
            self.expert_offset = 0
        self.num_experts = num_experts
        self.enable_tensor_parallel = enable_tensor_parallel and tp_size > 1

        self.experts = nn.ModuleList(
            [copy.deepcopy(expert) for _ in range(self.num_local_experts)]
        )
        for p in self.experts.parameters():
            setattr(p, "is_expert", True)

    def forward(self, inputs: torch.Tensor, dispatch_order: torch.Tensor, *args, **kwargs):
        # inputs: [B, S, H]; dispatch_order: [B*S] global expert index (top-1)
        shape = inputs.shape
        flat = inputs.reshape(-1, shape[-1])
        outputs = torch.zeros_like(flat)
        for local_idx, expert in enumerate(self.experts):
            global_idx = self.expert_offset + local_idx
            token_mask = dispatch_order.reshape(-1) == global_idx
            if token_mask.any():
                selected = flat[token_mask]
                expert_out = expert(selected, *args[1:], **kwargs)
                if isinstance(expert_out, tuple):
                    expert_out = expert_out[0]
                outputs[token_mask] = expert_out.to(outputs.dtype)
        if self.enable_tensor_parallel:
            outputs = _AllReduceCombine.apply(outputs, self.parallel_context)
        return outputs.reshape(shape)


class _AllReduceCombine(torch.autograd.Function):
    """Sum expert-sharded outputs across the TENSOR group (each token was
    computed on exactly one rank, zeros elsewhere)."""

    @staticmethod
    def forward(ctx, tensor, parallel_context):
        ctx.parallel_context = parallel_context
        tensor = tensor.contiguous()
        F.all_reduce(tensor, parallel_context=parallel_context,
                     parallel_mode=ParallelMode.TENSOR)
        return tensor

    @staticmethod
    def backward(ctx, grad):
        return grad, None
