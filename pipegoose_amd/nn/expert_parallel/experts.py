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
                 enable_tensor_parallel: bool, parallel_context: ParallelContext,
                 dispatch: str = "mask"):
        super().__init__()
        assert dispatch in ("mask", "alltoall")
        self.parallel_context = parallel_context
        self.dispatch = dispatch
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

        # share (don't copy) the ParallelContext a prototype module may hold:
        # ProcessGroups aren't picklable and must stay process-wide singletons
        memo = {id(parallel_context): parallel_context}
        self.experts = nn.ModuleList(
            [copy.deepcopy(expert, dict(memo)) for _ in range(self.num_local_experts)]
        )
        if self.enable_tensor_parallel:
            # An expert must be a DENSE local module: the EP axis reuses the
            # TENSOR group, so a TP-sharded expert would fire its internal
            # all-reduces with per-rank-different token counts → collective
            # mismatch.  (Native models build TP layers at construction;
            # pass a plain nn.Module via ExpertParallel(expert=...).)
            from pipegoose_amd.nn.tensor_parallel.linear import (
                ColumnParallelLinear, RowParallelLinear)
            for m in self.experts.modules():
                if isinstance(m, (ColumnParallelLinear, RowParallelLinear)):
                    raise TypeError(
                        "expert prototype contains tensor-parallel layers; "
                        "pass a dense expert module to ExpertParallel(expert=...)")
        for p in self.experts.parameters():
            setattr(p, "is_expert", True)
        # grouped-GEMM fast path: identical Sequential(Linear, act, Linear)
        # experts run as one batched MFMA GEMM over the expert-sorted tokens
        from pipegoose_amd.nn.expert_parallel.grouped import match_grouped_mlp
        self._grouped = match_grouped_mlp(self.experts)

    def forward(self, inputs: torch.Tensor, dispatch_order: torch.Tensor,
                weight: torch.Tensor = None, *args, **kwargs):
        # inputs: [B, S, H]; dispatch_order: [N] (top-1) or [N, k] global
        # expert ids; weight: optional [N, num_experts] routing weights
        # (capacity-dropped entries are 0).  With a weight matrix the output
        # is the gate-weighted combine (Switch semantics — this also gives
        # the router a gradient through the main path, which the reference's
        # unweighted scatter never did); without it, top-1 unweighted
        # (reference parity).
        shape = inputs.shape
        flat = inputs.reshape(-1, shape[-1])
        if self.dispatch == "alltoall" and self.enable_tensor_parallel:
            return self._forward_alltoall(flat, dispatch_order,
                                          weight).reshape(shape)
        k = 1 if dispatch_order.dim() == 1 else dispatch_order.size(-1)
        order2d = dispatch_order.reshape(-1, k)
        import os
        if self._grouped is not None and os.environ.get("PG_MOE_GROUPED") == "1":
            # measured SLOWER than the per-expert loop on the 1b7-MoE bench
            # (16.5k vs 41.3k tok/s): the ROCm transposed-bmm fault forces
            # contiguous weight copies every call (grouped.py), which
            # dominate.  Kept opt-in for re-evaluation when the backend bug
            # is fixed; the a2a path still uses the grouped bank (tokens/ep
            # per rank makes the copies proportionally cheaper there).
            outputs = self._forward_local_grouped(flat, order2d, weight)
            if self.enable_tensor_parallel:
                outputs = _AllReduceCombine.apply(outputs,
                                                  self.parallel_context)
            return outputs.reshape(shape)
        outputs = torch.zeros_like(flat)
        for local_idx, expert in enumerate(self.experts):
            global_idx = self.expert_offset + local_idx
            token_mask = (order2d == global_idx).any(dim=-1)
            if weight is not None:
                # capacity-dropped tokens (weight 0) are not computed
                token_mask = token_mask & (
                    weight.reshape(-1, self.num_experts)[:, global_idx] > 0)
            # A zero-token expert still runs (on a 0-row batch) so every
            # expert param gets a grad every step: DP ranks route different
            # tokens, and a rank that skipped an expert its EXPERT_DATA peer
            # ran would launch a different bucket-reduce sequence → hang.
            selected = flat[token_mask]
            # experts receive ONLY their tokens; block-level extras such
            # as HF Bloom's residual are handled by ExpertLayer
            expert_out = expert(selected)
            if isinstance(expert_out, tuple):
                expert_out = expert_out[0]
            expert_out = expert_out.to(outputs.dtype)
            if weight is not None:
                w = weight.reshape(-1, self.num_experts)[token_mask,
                                                         global_idx]
                expert_out = expert_out * w.unsqueeze(-1).to(outputs.dtype)
                idx = token_mask.nonzero(as_tuple=True)[0]
                outputs = outputs.index_put((idx,), expert_out,
                                            accumulate=True)  # top-2 sums
            else:
                outputs[token_mask] = expert_out
        if self.enable_tensor_parallel:
            outputs = _AllReduceCombine.apply(outputs, self.parallel_context)
        return outputs.reshape(shape)

    def _forward_local_grouped(self, flat: torch.Tensor,
                               order2d: torch.Tensor,
                               weight: torch.Tensor = None) -> torch.Tensor:
        """Mask-dispatch semantics via one grouped GEMM over the tokens
        routed to THIS rank's experts (all of them when EP is off)."""
        from pipegoose_amd.nn.expert_parallel.grouped import (
            grouped_mlp_forward)
        N, k = order2d.shape
        dev = flat.device
        tok = torch.arange(N, device=dev).repeat_interleave(k)
        ridx = order2d.reshape(-1)
        keep = (ridx >= self.expert_offset) & \
               (ridx < self.expert_offset + self.num_local_experts)
        if weight is not None:
            wvals = weight.reshape(N, self.num_experts)[tok, ridx]
            keep = keep & (wvals > 0)
        tok, ridx = tok[keep], ridx[keep] - self.expert_offset
        perm = torch.argsort(ridx, stable=True)
        tok, ridx = tok[perm], ridx[perm]
        counts = torch.bincount(ridx,
                                minlength=self.num_local_experts).tolist()
        w1, b1, w2, b2, act = self._grouped
        expert_out = grouped_mlp_forward(flat[tok], counts, w1, b1, w2, b2,
                                         act).to(flat.dtype)
        if weight is not None:
            expert_out = expert_out * wvals[keep][perm].unsqueeze(-1) \
                .to(flat.dtype)
        if k == 1:
            # top-1: each token appears at most once — plain scatter
            outputs = torch.zeros_like(flat)
            return outputs.index_put((tok,), expert_out)
        # top-k: a token can receive k contributions; accumulate in fp32
        # (bf16 scattered atomics are neither precise nor reliable here)
        out32 = torch.zeros(flat.shape, device=flat.device,
                            dtype=torch.float32)
        out32 = out32.index_put((tok,), expert_out.float(), accumulate=True)
        return out32.to(flat.dtype)

    def _forward_alltoall(self, flat: torch.Tensor,
                          dispatch_order: torch.Tensor,
                          weight: torch.Tensor = None):
        """EP dispatch over xGMI.  The activation is replicated across the EP
        group, so first each rank takes its 1/ep chunk of the token set (no
        redundant expert FLOPs — the mask path recomputes every token on every
        rank), all-to-alls its chunk to the expert-owner ranks, computes, and
        all-to-alls back; a final dim-0 all-gather rebuilds the replicated
        output.  Backward mirrors: chunk the output grad, reverse exchanges,
        all-gather input grads — every rank ends with the FULL input gradient
        (the mask path leaves per-rank partials that are only correct after a
        downstream TP sum).

        With a routing-weight matrix: capacity-dropped tokens are NOT put on
        the wire (their output row stays 0 — the residual path carries them,
        Switch semantics), top-2 tokens ship once per selected expert, and
        the combine is gate-weighted (giving the router main-path gradient).
        """
        from pipegoose_amd.nn.expert_parallel.dispatch import AllToAllDispatcher
        from pipegoose_amd.nn.tensor_parallel._functional import (
            _Gather, _Scatter)
        if not hasattr(self, "_dispatcher"):
            self._dispatcher = AllToAllDispatcher(
                self.num_experts, self.parallel_context, ParallelMode.TENSOR)
        ep = self.parallel_context.get_world_size(ParallelMode.TENSOR)
        rank = self.parallel_context.get_local_rank(ParallelMode.TENSOR)
        assert flat.size(0) % ep == 0, \
            f"token count {flat.size(0)} not divisible by ep={ep}"
        k = 1 if dispatch_order.dim() == 1 else dispatch_order.size(-1)
        chunk = _Scatter.apply(flat, 0, self.parallel_context)
        n = chunk.size(0)
        route_chunk = dispatch_order.reshape(-1, k).chunk(ep, dim=0)[rank]

        if weight is None and k == 1:
            send_tokens, send_routes = chunk, route_chunk.reshape(-1)
            tok_idx = None
        else:
            w_chunk = weight.reshape(-1, self.num_experts).chunk(ep, dim=0)[rank]
            tok_rep = torch.arange(n, device=chunk.device) \
                .repeat_interleave(k)
            ridx = route_chunk.reshape(-1)
            wvals = w_chunk[tok_rep, ridx]
            keep = wvals > 0  # capacity-dropped entries never hit the wire
            tok_idx = tok_rep[keep]
            send_tokens = chunk[tok_idx]
            send_routes = ridx[keep]
            wkeep = wvals[keep]

        recv, local_idx, state = self._dispatcher.dispatch(send_tokens,
                                                           send_routes)
        counts = torch.bincount(local_idx, minlength=self.num_local_experts).tolist()
        if self._grouped is not None and recv.size(0) > 0:
            from pipegoose_amd.nn.expert_parallel.grouped import (
                grouped_mlp_forward)
            w1, b1, w2, b2, act = self._grouped
            expert_out = grouped_mlp_forward(recv, counts, w1, b1, w2, b2, act)
        else:
            outs = []
            start = 0
            for i, expert in enumerate(self.experts):
                seg = recv[start:start + counts[i]]
                out = expert(seg)
                if isinstance(out, tuple):
                    out = out[0]
                outs.append(out.to(recv.dtype))
                start += counts[i]
            expert_out = torch.cat(outs, dim=0) if outs else recv
        combined = self._dispatcher.combine(expert_out, state)
        if tok_idx is not None:
            combined = combined * wkeep.unsqueeze(-1).to(combined.dtype)
            if k == 1:  # no per-token overlap: plain scatter
                outc = torch.zeros_like(chunk)
                combined = outc.index_put((tok_idx,), combined)
            else:       # top-k sums accumulate in fp32 (see grouped path)
                out32 = torch.zeros(chunk.shape, device=chunk.device,
                                    dtype=torch.float32)
                out32 = out32.index_put((tok_idx,), combined.float(),
                                        accumulate=True)
                combined = out32.to(chunk.dtype)
        return _Gather.apply(combined, 0, self.parallel_context)


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
