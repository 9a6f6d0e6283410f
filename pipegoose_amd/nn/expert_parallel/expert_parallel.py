"""ExpertParallel: replace selected MLP blocks of a transformer with MoE
ExpertLayers (reference: nn/expert_parallel/expert_parallel.py:53-79)."""
import re
from typing import Callable, List, Optional, Union

from torch import nn

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.expert_parallel.layers import ExpertLayer
from pipegoose_amd.nn.parallel import Parallel


class ExpertParallel(Parallel):
    def __init__(
        self,
        module: nn.Module,
        num_experts: int,
        expert: Optional[nn.Module] = None,
        mapping: Optional[List[int]] = None,
        router: Union[int, Callable] = 1,
        noise_policy: Union[str, Callable] = "gaussian",
        enable_tensor_parallel: bool = False,
        parallel_context: ParallelContext = None,
        dispatch: str = "mask",
    ):
        super().__init__(module, parallel_context)
        self.dispatch = dispatch
        tp_size = parallel_context.get_world_size(ParallelMode.TENSOR)
        if enable_tensor_parallel:
            assert num_experts % tp_size == 0, (
                f"num_experts ({num_experts}) must be divisible by tp ({tp_size})"
            )
        self.num_experts = num_experts
        self.expert = expert
        self.mapping = mapping
        self.router = router
        self.enable_tensor_parallel = enable_tensor_parallel

    def parallelize(self) -> nn.Module:
        pattern = re.compile(r"^(?:transformer\.)?(?:h|layers)\.(\d+)\.mlp$", re.IGNORECASE)
        replaced = 0
        for name, module in list(self.module.named_modules()):
            m = pattern.search(name)
            if m is None:
                continue
            layer_idx = int(m.group(1))
            if self.mapping is not None and layer_idx not in self.mapping:
                continue
            expert = self.expert if self.expert is not None else module
            expert_layer = ExpertLayer(
                self.num_experts, expert, self.router,
                self.enable_tensor_parallel, self.parallel_context,
                dispatch=self.dispatch,
            )
            _set_submodule(self.module, name, expert_layer)
            replaced += 1
        assert replaced > 0, "no MLP blocks matched for expert replacement"
        self._save_metadata(self.module, self.parallel_context)
        return self.module


def _set_submodule(model: nn.Module, dotted: str, new_module: nn.Module):
    parts = dotted.split(".")
    parent = model
    for p in parts[:-1]:
        parent = getattr(parent, p)
    setattr(parent, parts[-1], new_module)
