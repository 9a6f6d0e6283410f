"""TensorParallel: drop-in TP surgery on an HF (or native) model.

Reference parity: nn/tensor_parallel/tensor_parallel.py:27-71.
Walks leaf modules (skipping ExpertLayer subtrees, which belong to
ExpertParallel), matches each against the parallelizer registry, mutates the
model in place.
"""
import torch
from torch import nn

from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.parallel import Parallel
from pipegoose_amd.nn.tensor_parallel.parallelizer import (
    Conv1DParallelizer,
    EmbeddingParallelizer,
    LayerNormParallelizer,
    LinearParallelizer,
    LMHeadParallelizer,
)

PARALLELIZERS = [
    EmbeddingParallelizer,
    LMHeadParallelizer,   # before LinearParallelizer: lm_head is also a Linear
    LinearParallelizer,
    Conv1DParallelizer,   # GPT-2 family (transposed-weight Conv1D)
    LayerNormParallelizer,
]


class TensorParallel(Parallel):
    def parallelize(self) -> nn.Module:
        module = self.module
        if self.parallel_context.get_world_size(ParallelMode.TENSOR) > 1:
            leaves = self._get_leaf_modules(module)
            for name, leaf in leaves:
                parallelizer = self._find_parallelizer(name, leaf)
                if parallelizer is not None:
                    parallelizer(name, leaf, module, self.parallel_context).parallelize()
            self._adapt_attention_modules(module)
        self._save_metadata(module, self.parallel_context)
        return module

    def _adapt_attention_modules(self, model: nn.Module):
        """HF attention modules keep full head counts after the weight split
        (the reference never fixed this — its HF-TP path crashed on the
        fused-qkv reshape).  Divide per-module head attributes by tp when the
        module owns a column-split child, and apply per-architecture quirks
        (BloomAttention: the model-level alibi tensor arrives with ALL heads;
        slice this rank's head block)."""
        from pipegoose_amd.nn.tensor_parallel.linear import ColumnParallelLinear
        tp = self.parallel_context.get_world_size(ParallelMode.TENSOR)
        rank = self.parallel_context.get_local_rank(ParallelMode.TENSOR)
        for mod in model.modules():
            has_cpl = any(isinstance(c, ColumnParallelLinear)
                          for c in mod.children())
            if not has_cpl:
                continue
            restored = {}
            for attr in ("num_heads", "num_attention_heads",
                         "num_key_value_heads", "split_size", "all_head_size"):
                val = getattr(mod, attr, None)
                if isinstance(val, int) and val % tp == 0 and val >= tp:
                    restored[attr] = val
                    setattr(mod, attr, val // tp)
            if restored:
                mod._tp_head_attrs = restored
            if type(mod).__name__ == "BloomAttention":
                mod._tp_orig_forward = mod.forward
                _patch_bloom_attention_alibi(mod, rank, tp)

    @staticmethod
    def _get_leaf_modules(model: nn.Module):
        from pipegoose_amd.nn.expert_parallel.layers import ExpertLayer
        leaves = []
        expert_prefixes = [name for name, m in model.named_modules()
                           if isinstance(m, ExpertLayer)]
        for name, m in model.named_modules():
            if list(m.children()):
                continue
            if any(name.startswith(p) for p in expert_prefixes):
                continue
            leaves.append((name, m))
        return leaves

    @staticmethod
    def _find_parallelizer(module_name: str, module: nn.Module):
        for p in PARALLELIZERS:
            if p.is_parallelizable(module_name, module):
                return p
        return None

    def deparallelize(self) -> nn.Module:
        """Reverse the surgery: all-gather each sliced weight over the
        TENSOR group, restore the original module classes and attention
        head counts.  The reference declares this per-parallelizer but
        every body is ``pass`` (parallelizer.py:57-228); here the model
        round-trips to its full single-rank form (state_dict equal to the
        pre-parallelize one up to collective ordering)."""
        import torch.distributed as dist

        module = self.module
        pc = self.parallel_context
        tp = pc.get_world_size(ParallelMode.TENSOR)
        if tp == 1:
            return module
        group = pc.get_group(ParallelMode.TENSOR)

        def gather(t: torch.Tensor, dim: int) -> torch.Tensor:
            parts = [torch.empty_like(t) for _ in range(tp)]
            dist.all_gather(parts, t.contiguous(), group=group)
            return torch.cat(parts, dim=dim)

        for mod in module.modules():
            ha = getattr(mod, "_tp_head_attrs", None)
            if ha:
                for attr, val in ha.items():
                    setattr(mod, attr, val)
                del mod._tp_head_attrs
            if hasattr(mod, "_tp_orig_forward"):
                mod.forward = mod._tp_orig_forward
                del mod._tp_orig_forward

        for name, m in module.named_modules():
            info = getattr(m, "_tp_info", None)
            if info is None:
                continue
            kind = info["kind"]
            conv1d = info.get("conv1d", False)
            if kind == "column":
                if not info.get("tied_weight", False):
                    m.weight.data = gather(m.weight.data, 0)
                if m.bias is not None:
                    m.bias.data = gather(m.bias.data, 0)
            elif kind == "fused_column":
                n = info["fused_n"]
                o_nt, in_f = m.weight.shape[0] // n, m.weight.shape[1]
                parts = [torch.empty_like(m.weight.data) for _ in range(tp)]
                dist.all_gather(parts, m.weight.data.contiguous(), group=group)
                m.weight.data = torch.cat(
                    [torch.cat([p.view(n, o_nt, in_f)[i] for p in parts], 0)
                     for i in range(n)], 0)
                if m.bias is not None:
                    bparts = [torch.empty_like(m.bias.data) for _ in range(tp)]
                    dist.all_gather(bparts, m.bias.data.contiguous(), group=group)
                    m.bias.data = torch.cat(
                        [torch.cat([p.view(n, o_nt)[i] for p in bparts], 0)
                         for i in range(n)], 0)
            elif kind == "row":
                m.weight.data = gather(m.weight.data, 1)
            elif kind == "embedding":
                m.weight.data = gather(m.weight.data, 0)
                for attr in ("partition_size", "vocab_start_idx",
                             "vocab_end_idx"):
                    if hasattr(m, attr):
                        delattr(m, attr)
            # layer_norm: class swap only

            if conv1d and kind in ("column", "fused_column", "row"):
                m.weight.data = m.weight.data.t().contiguous()
            m.__class__ = info["orig_cls"]
            if kind in ("column", "fused_column") and not conv1d:
                m.out_features = m.weight.shape[0]
            elif kind == "row" and not conv1d:
                m.in_features = m.weight.shape[1]
            for attr in ("gather_output", "sequence_parallel",
                         "parallel_context"):
                if hasattr(m, attr):
                    try:
                        delattr(m, attr)
                    except AttributeError:
                        pass
            del m._tp_info
        return module


def _patch_bloom_attention_alibi(attn, tp_rank: int, tp_size: int):
    """HF BloomModel builds one alibi tensor [B * n_head_full, 1, S] and hands
    it to every (now head-sharded) attention; wrap forward to slice out this
    rank's contiguous head block."""
    orig_forward = attn.forward
    local_heads = attn.num_heads  # already divided

    def forward(*args, **kwargs):
        def slice_alibi(alibi, batch_hint):
            full = alibi.size(0) // max(batch_hint, 1)
            if full == local_heads:
                return alibi
            B = alibi.size(0) // (local_heads * tp_size)
            a = alibi.view(B, local_heads * tp_size, *alibi.shape[1:])
            a = a[:, tp_rank * local_heads:(tp_rank + 1) * local_heads]
            return a.reshape(B * local_heads, *alibi.shape[1:])

        hidden = kwargs.get("hidden_states", args[0] if args else None)
        bsz = hidden.size(0) if hidden is not None else 1
        if "alibi" in kwargs and torch.is_tensor(kwargs["alibi"]):
            kwargs["alibi"] = slice_alibi(kwargs["alibi"], bsz)
        elif len(args) >= 3 and torch.is_tensor(args[2]):
            args = list(args)
            args[2] = slice_alibi(args[2], bsz)
        return orig_forward(*args, **kwargs)

    attn.forward = forward
