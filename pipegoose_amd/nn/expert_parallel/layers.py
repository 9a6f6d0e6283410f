"""MoE layer: router -> loss bookkeeping -> expert dispatch.

Reference parity: nn/expert_parallel/layers.py:40-48.
"""
from torch import nn

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.nn.expert_parallel.expert_context import ExpertContext


class ExpertLayer(nn.Module):
    def __init__(self, num_experts: int, expert: nn.Module, router: nn.Module,
                 enable_tensor_parallel: bool, parallel_context: ParallelContext,
                 dispatch: str = "mask"):
        super().__init__()
        from pipegoose_amd.nn.expert_parallel.experts import Experts
        self.router = router
        self._experts = Experts(num_experts, expert, enable_tensor_parallel,
                                parallel_context, dispatch=dispatch)
        self.parallel_context = parallel_context

    @property
    def experts(self):
        return self._experts.experts

    def forward(self, *args, **kwargs):
        inputs = args[0]
        router_output = self.router(inputs)
        expert_context = ExpertContext.get_instance()
        expert_context.push_aux_loss(router_output.aux_loss)
        expert_context.push_z_loss(router_output.z_loss)
        outputs = self._experts(inputs, router_output.dispatch_order,
                                router_output.weight)
        # HF Bloom's MLP signature is (hidden, residual) with the residual
        # added inside the block; when this layer replaces such an MLP the
        # residual arrives as the 2nd positional arg — add it here.
        import torch as _torch
        if len(args) >= 2 and _torch.is_tensor(args[1]) \
                and args[1].shape == outputs.shape:
            outputs = outputs + args[1]
        return outputs
