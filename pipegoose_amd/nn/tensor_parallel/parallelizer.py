"""In-place module surgery: swap nn.Linear/Embedding/LayerNorm leaves for their
tensor-parallel counterparts and slice the weights.

Reference parity: nn/tensor_parallel/parallelizer.py (class-swap via
``module.__class__ = ...`` + torch.chunk slicing).  Differences:
  - no vocab padding mismatch between embedding and LM head (reference bug,
    parallelizer.py:153-169): both require divisibility and share the slice;
  - ``bias=False`` handled throughout.
"""
from abc import ABC, abstractmethod

import torch
from torch import nn

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.tensor_parallel.embedding import ParallelEmbedding
from pipegoose_amd.nn.tensor_parallel.layer_norm import LayerNorm
from pipegoose_amd.nn.tensor_parallel.linear import ColumnParallelLinear, RowParallelLinear
from pipegoose_amd.nn.tensor_parallel.parallel_mapping import TensorParallelMapping


def get_partition(tensor: torch.Tensor, dim: int, parallel_context: ParallelContext) -> torch.Tensor:
    world = parallel_context.get_world_size(ParallelMode.TENSOR)
    rank = parallel_context.get_local_rank(ParallelMode.TENSOR)
    return tensor.chunk(world, dim=dim)[rank].contiguous()


class ModuleParallelizer(ABC):
    def __init__(self, module_name: str, module: nn.Module, model: nn.Module,
                 parallel_context: ParallelContext):
        self.module_name = module_name
        self.module = module
        self.model = model
        self.parallel_context = parallel_context

    @staticmethod
    @abstractmethod
    def is_parallelizable(module_name: str, module: nn.Module) -> bool:
        ...

    @abstractmethod
    def parallelize(self) -> nn.Module:
        ...


class LinearParallelizer(ModuleParallelizer):
    @staticmethod
    def is_parallelizable(module_name, module):
        return isinstance(module, nn.Linear) and (
            TensorParallelMapping.is_column_parallel(module_name)
            or TensorParallelMapping.is_row_parallel(module_name)
        )

    def parallelize(self):
        module, pc = self.module, self.parallel_context
        orig_cls = type(module)
        if TensorParallelMapping.is_column_parallel(self.module_name):
            module.__class__ = ColumnParallelLinear
            module.weight.data = get_partition(module.weight.data, dim=0, parallel_context=pc)
            if module.bias is not None:
                module.bias.data = get_partition(module.bias.data, dim=0, parallel_context=pc)
            module.gather_output = False
            module.out_features = module.weight.shape[0]
            module._tp_info = {"kind": "column", "orig_cls": orig_cls}
        else:
            module.__class__ = RowParallelLinear
            module.weight.data = get_partition(module.weight.data, dim=1, parallel_context=pc)
            module.in_features = module.weight.shape[1]
            module._tp_info = {"kind": "row", "orig_cls": orig_cls}
        module.parallel_context = pc
        return module


class Conv1DParallelizer(ModuleParallelizer):
    """HF GPT-2 ``Conv1D`` (weight stored [in, out] — transposed vs
    nn.Linear).  Converted in place to Column/RowParallelLinear with the
    weight transposed and sliced.  Fused ``c_attn`` slices each of the n
    stacked blocks separately: GPT-2 lays q|k|v out BLOCKWISE along the
    output dim, so a plain chunk would give rank 0 all of Q (the reference
    never supported Conv1D models at all)."""

    @staticmethod
    def is_parallelizable(module_name, module):
        return (type(module).__name__ == "Conv1D" and hasattr(module, "nf")
                and (TensorParallelMapping.is_column_parallel(module_name)
                     or TensorParallelMapping.is_row_parallel(module_name)
                     or TensorParallelMapping.is_fused_column(module_name)))

    def parallelize(self):
        module, pc = self.module, self.parallel_context
        w = module.weight.data.t().contiguous()   # -> [out, in]
        b = module.bias.data if getattr(module, "bias", None) is not None \
            else None
        orig_cls = type(module)
        if TensorParallelMapping.is_fused_column(self.module_name):
            n = TensorParallelMapping.get_fused_count(self.module_name)
            out = w.shape[0]
            assert out % n == 0
            blocks = w.view(n, out // n, w.shape[1])
            module.weight = nn.Parameter(torch.cat(
                [get_partition(blk, 0, pc) for blk in blocks], dim=0))
            if b is not None:
                bb = b.view(n, out // n)
                module.bias = nn.Parameter(torch.cat(
                    [get_partition(x, 0, pc) for x in bb], dim=0))
            module.__class__ = ColumnParallelLinear
            module.gather_output = False
            module.out_features = module.weight.shape[0]
            module._tp_info = {"kind": "fused_column", "orig_cls": orig_cls,
                               "conv1d": True, "fused_n": n}
        elif TensorParallelMapping.is_column_parallel(self.module_name):
            module.weight = nn.Parameter(get_partition(w, 0, pc))
            if b is not None:
                module.bias = nn.Parameter(get_partition(b, 0, pc))
            module.__class__ = ColumnParallelLinear
            module.gather_output = False
            module.out_features = module.weight.shape[0]
            module._tp_info = {"kind": "column", "orig_cls": orig_cls,
                               "conv1d": True}
        else:  # row-parallel: split the input dim, bias kept whole
            module.weight = nn.Parameter(get_partition(w, 1, pc))
            if b is not None:
                module.bias = nn.Parameter(b.clone())
            module.__class__ = RowParallelLinear
            module.in_features = module.weight.shape[1]
            module._tp_info = {"kind": "row", "orig_cls": orig_cls,
                               "conv1d": True}
        module.parallel_context = pc
        return module


class EmbeddingParallelizer(ModuleParallelizer):
    @staticmethod
    def is_parallelizable(module_name, module):
        return isinstance(module, nn.Embedding)

    def parallelize(self):
        module, pc = self.module, self.parallel_context
        world = pc.get_world_size(ParallelMode.TENSOR)
        rank = pc.get_local_rank(ParallelMode.TENSOR)
        vocab_size = module.weight.shape[0]
        assert vocab_size % world == 0, (
            f"vocab size {vocab_size} must be divisible by tp={world}"
        )
        partition = vocab_size // world
        orig_cls = type(module)
        module.weight.data = get_partition(module.weight.data, dim=0, parallel_context=pc)
        module.__class__ = ParallelEmbedding
        module.num_embeddings = vocab_size
        module.partition_size = partition
        module.vocab_start_idx = rank * partition
        module.vocab_end_idx = (rank + 1) * partition
        module.parallel_context = pc
        module._tp_info = {"kind": "embedding", "orig_cls": orig_cls}
        return module


class LayerNormParallelizer(ModuleParallelizer):
    """LayerNorm stays full-width; swapped so it runs the fused HIP kernel."""

    @staticmethod
    def is_parallelizable(module_name, module):
        return isinstance(module, nn.LayerNorm)

    def parallelize(self):
        module, pc = self.module, self.parallel_context
        orig_cls = type(module)
        normalized_shape = tuple(module.normalized_shape)
        module.__class__ = LayerNorm
        module.normalized_shape = normalized_shape
        module.parallel_context = pc
        module._tp_info = {"kind": "layer_norm", "orig_cls": orig_cls}
        return module


class LMHeadParallelizer(ModuleParallelizer):
    """Column-split LM head.  With weights tied to the (already-sliced)
    embedding the slice is skipped (reference: parallelizer.py:194-229)."""

    @staticmethod
    def is_parallelizable(module_name, module):
        return isinstance(module, nn.Linear) and TensorParallelMapping.is_lm_head(module_name)

    def parallelize(self):
        module, pc = self.module, self.parallel_context
        orig_cls = type(module)
        module.__class__ = ColumnParallelLinear
        embed_weight = _get_tied_embedding_weight(self.model)
        tied = embed_weight is not None and module.weight is embed_weight
        if not tied:
            module.weight.data = get_partition(module.weight.data, dim=0, parallel_context=pc)
        if getattr(module, "bias", None) is not None:
            module.bias.data = get_partition(module.bias.data, dim=0, parallel_context=pc)
        module.gather_output = True  # logits gathered for the (parallel) loss
        module.out_features = module.weight.shape[0]
        module.parallel_context = pc
        module._tp_info = {"kind": "column", "orig_cls": orig_cls,
                           "tied_weight": tied}
        return module


def _get_tied_embedding_weight(model: nn.Module):
    for m in model.modules():
        if isinstance(m, (nn.Embedding, ParallelEmbedding)):
            return m.weight
    return None
