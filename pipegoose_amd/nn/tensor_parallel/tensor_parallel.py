"""TensorParallel: drop-in TP surgery on an HF (or native) model.

Reference parity: nn/tensor_parallel/tensor_parallel.py:27-71.
Walks leaf modules (skipping ExpertLayer subtrees, which belong to
ExpertParallel), matches each against the parallelizer registry, mutates the
model in place.
"""
from torch import nn

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.parallel import Parallel
from pipegoose_amd.nn.tensor_parallel.parallelizer import (
    EmbeddingParallelizer,
    LayerNormParallelizer,
    LinearParallelizer,
    LMHeadParallelizer,
)

PARALLELIZERS = [
    EmbeddingParallelizer,
    LMHeadParallelizer,   # before LinearParallelizer: lm_head is also a Linear
    LinearParallelizer,
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
        self._save_metadata(module, self.parallel_context)
        return module

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
