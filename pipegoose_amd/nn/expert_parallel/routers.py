"""Switch-Transformer top-k routers (reference: nn/expert_parallel/routers.py).

Gate GEMM in fp32, multiplicative jitter noise, top-k mask, Switch aux
load-balance loss, ST-MoE router z-loss, optional expert-capacity truncation.
On GPU the softmax+top-k+capacity pass runs in a fused HIP kernel
(pipegoose_amd.ops.router) for k in {1, 2}.
"""
from dataclasses import dataclass
from typing import Optional, Tuple

import torch
from torch import nn
import torch.nn.functional as TF


@dataclass
class RouterOutput:
    dispatch_order: torch.Tensor      # [N] expert index per token (top-1) or [N,k]
    weight: torch.Tensor              # routing weights [N, num_experts] (masked)
    aux_loss: torch.Tensor
    z_loss: torch.Tensor


class SwitchNoisePolicy:
    """Multiplicative jitter noise in [1-eps, 1+eps).

    Draws come from a dedicated generator seeded identically on every rank:
    under tensor parallelism the default CUDA seeds are deliberately
    TP-decorrelated (Megatron dropout convention), but ROUTING must agree
    across the expert group — with divergent noise the mask-dispatch path
    double- or zero-counts tokens."""

    def __init__(self, eps: float = 0.1, seed: int = 1234):
        self.eps = eps
        self.seed = seed
        self._gens = {}

    def _gen(self, device: torch.device) -> torch.Generator:
        key = (device.type, device.index)
        g = self._gens.get(key)
        if g is None:
            g = torch.Generator(device=device)
            g.manual_seed(self.seed)
            self._gens[key] = g
        return g

    def sample_like(self, logits: torch.Tensor) -> torch.Tensor:
        noise = torch.rand(logits.shape, device=logits.device,
                           dtype=logits.dtype, generator=self._gen(logits.device))
        return 1.0 - self.eps + 2.0 * self.eps * noise


class _TopKRouter(nn.Module):
    def __init__(
        self,
        noise_policy: SwitchNoisePolicy,
        top_k: int,
        num_experts: int,
        d_model: int,
        expert_capacity: Optional[Tuple[float, float]] = None,
        alpha: float = 0.01,
        eps: float = 0.1,
    ):
        super().__init__()
        self.noise_policy = noise_policy
        self.top_k = top_k
        self.num_experts = num_experts
        self.expert_capacity = expert_capacity
        self.alpha = alpha
        self.eps = eps
        self.gate = nn.Linear(d_model, num_experts)

    def _expert_capacity(self, total_tokens: int) -> int:
        capacity_factor = self.expert_capacity[0 if self.training else 1]
        return int(capacity_factor * total_tokens / self.num_experts)

    def _aux_loss(self, router_probs: torch.Tensor, expert_mask: torch.Tensor) -> torch.Tensor:
        # Switch load-balance loss: num_experts * sum(f_i * P_i)
        fraction_tokens = expert_mask.float().mean(dim=0)
        fraction_probs = router_probs.mean(dim=0)
        return self.num_experts * torch.sum(fraction_tokens * fraction_probs)

    def _z_loss(self, router_logits: torch.Tensor) -> torch.Tensor:
        return torch.logsumexp(router_logits, dim=-1).square().mean()

    def forward(self, inputs: torch.Tensor) -> RouterOutput:
        # inputs: [B, S, H] or [N, H]
        orig_dtype = inputs.dtype
        hidden = inputs.reshape(-1, inputs.size(-1)).float()
        gate_bias = self.gate.bias.float() if self.gate.bias is not None else None
        logits = TF.linear(hidden, self.gate.weight.float(), gate_bias)
        if self.training:
            logits = logits * self.noise_policy.sample_like(logits)
        elif (logits.is_cuda and not torch.is_grad_enabled()
              and self.top_k == 1 and self.num_experts <= 64):
            out = self._forward_fused(logits, orig_dtype)
            if out is not None:
                return out

        probs = TF.softmax(logits, dim=-1)
        topk_val, topk_idx = probs.topk(self.top_k, dim=-1)

        expert_mask = torch.zeros_like(probs)
        expert_mask.scatter_(-1, topk_idx, 1.0)

        aux_loss = self._aux_loss(probs, expert_mask.bool())
        z_loss = self._z_loss(logits)

        if self.expert_capacity is not None:
            capacity = self._expert_capacity(hidden.size(0))
            # position of each token within its expert's queue (top-1 column only)
            position_in_expert = torch.cumsum(expert_mask, dim=0) * expert_mask
            within_capacity = (position_in_expert <= capacity).float()
            expert_mask = expert_mask * within_capacity

        weight = probs * expert_mask
        dispatch_order = topk_idx.squeeze(-1) if self.top_k == 1 else topk_idx
        return RouterOutput(
            dispatch_order=dispatch_order,
            weight=weight.to(orig_dtype),
            aux_loss=aux_loss,
            z_loss=z_loss,
        )


    def _forward_fused(self, logits: torch.Tensor, orig_dtype):
        """Serving path: one fused HIP pass (softmax+top1+colsum+lse,
        ops/csrc/router.hip) instead of five elementwise/reduce kernels.
        Training keeps the autograd torch ops (loss grads)."""
        from pipegoose_amd.ops import get_extension
        ext = get_extension()
        if ext is None or not hasattr(ext, "router_topk"):
            return None
        N = logits.size(0)
        idx, val, colsum, count, lse = ext.router_topk(logits.contiguous(), 1)
        idx = idx.squeeze(-1).long()
        weight = torch.zeros_like(logits)
        weight.scatter_(-1, idx.unsqueeze(-1), val)
        aux = self.num_experts * torch.sum(
            (count.float() / N) * (colsum / N))
        z = lse.square().mean()
        if self.expert_capacity is not None:
            capacity = self._expert_capacity(N)
            mask = torch.zeros_like(logits)
            mask.scatter_(-1, idx.unsqueeze(-1), 1.0)
            position = torch.cumsum(mask, dim=0) * mask
            weight = weight * (position <= capacity).float()
        return RouterOutput(dispatch_order=idx, weight=weight.to(orig_dtype),
                            aux_loss=aux, z_loss=z)


class Top1Router(_TopKRouter):
    def __init__(self, noise_policy: SwitchNoisePolicy, num_experts: int, d_model: int,
                 expert_capacity: Optional[Tuple[float, float]] = None, **kwargs):
        super().__init__(noise_policy, 1, num_experts, d_model, expert_capacity, **kwargs)


class Top2Router(_TopKRouter):
    def __init__(self, noise_policy: SwitchNoisePolicy, num_experts: int, d_model: int,
                 expert_capacity: Optional[Tuple[float, float]] = None, **kwargs):
        super().__init__(noise_policy, 2, num_experts, d_model, expert_capacity, **kwargs)
