"""ParallelContext: rank/group bookkeeping for the 4D process grid.

API mirrors the reference (pipegoose/distributed/parallel_context.py) so user
code is drop-in; the implementation is MI355X-first:

  - one process per GPU, groups built on the ``nccl`` backend (RCCL over xGMI
    on ROCm) when GPUs are present, ``gloo`` on CPU;
  - no TensorPipe RPC: pipeline data moves over RCCL P2P send/recv
    (see nn/pipeline_parallel/), the control plane is collective-based;
  - device binding via ``torch.cuda.set_device`` (HIP).
"""
import os
import random
from typing import List, Optional

import numpy as np
import torch
import torch.distributed as dist

from pipegoose_amd.distributed._initializers import (
    ContextParallelGroupInitializer,
    DataParallelGroupInitializer,
    ExpertDataParallelGroupInitializer,
    PipelineParallelGroupInitializer,
    TensorParallelGroupInitializer,
)
from pipegoose_amd.distributed.parallel_mode import ParallelMode

_GLOBAL_CONTEXT: Optional["ParallelContext"] = None

DEFAULT_SEED = 69


class ParallelContext:
    """Registry of ranks, process groups and world sizes per ParallelMode."""

    @staticmethod
    def from_torch(
        tensor_parallel_size: int = 1,
        pipeline_parallel_size: int = 1,
        data_parallel_size: int = 1,
        context_parallel_size: int = 1,
        seed: int = DEFAULT_SEED,
        backend: Optional[str] = None,
    ) -> "ParallelContext":
        """Bootstrap from torchrun env vars (RANK/LOCAL_RANK/WORLD_SIZE/MASTER_*)."""
        rank = int(os.environ.get("RANK", 0))
        local_rank = int(os.environ.get("LOCAL_RANK", 0))
        world_size = int(os.environ.get("WORLD_SIZE", 1))
        host = os.environ.get("MASTER_ADDR", "127.0.0.1")
        port = int(os.environ.get("MASTER_PORT", 29500))
        return ParallelContext(
            rank=rank,
            local_rank=local_rank,
            world_size=world_size,
            local_world_size=int(os.environ.get("LOCAL_WORLD_SIZE", world_size)),
            host=host,
            port=port,
            backend=backend,
            seed=seed,
            tensor_parallel_size=tensor_parallel_size,
            pipeline_parallel_size=pipeline_parallel_size,
            data_parallel_size=data_parallel_size,
            context_parallel_size=context_parallel_size,
        )

    def __init__(
        self,
        rank: int,
        local_rank: int,
        world_size: int,
        local_world_size: int,
        host: str,
        port: int,
        backend: Optional[str],
        seed: int,
        tensor_parallel_size: int,
        pipeline_parallel_size: int,
        data_parallel_size: Optional[int] = None,
        context_parallel_size: int = 1,
    ):
        if data_parallel_size is None:
            data_parallel_size = world_size // (
                tensor_parallel_size * pipeline_parallel_size
                * context_parallel_size)
        assert world_size == (tensor_parallel_size * pipeline_parallel_size
                              * data_parallel_size * context_parallel_size), (
            f"world_size ({world_size}) != tp ({tensor_parallel_size}) x pp "
            f"({pipeline_parallel_size}) x dp ({data_parallel_size}) x cp "
            f"({context_parallel_size})"
        )

        self.tensor_parallel_size = tensor_parallel_size
        self.pipeline_parallel_size = pipeline_parallel_size
        self.data_parallel_size = data_parallel_size
        self.context_parallel_size = context_parallel_size

        self._rank = rank
        self._local_rank = local_rank
        self._world_size = world_size
        self._local_world_size = local_world_size

        self._groups = {}
        self._ranks_in_group = {}
        self._local_ranks = {}
        self._world_sizes = {}

        if backend is None:
            backend = "nccl" if torch.cuda.is_available() else "gloo"
        self.backend = backend

        self.init_global_dist(rank, world_size, backend, host, port)
        if torch.cuda.is_available():
            torch.cuda.set_device(local_rank)
        self.init_parallel_groups()

        self.set_seed(seed)
        self._set_context()

    # ------------------------------------------------------------------ setup

    def init_global_dist(self, rank, world_size, backend, host, port):
        if not dist.is_initialized():
            init_method = f"tcp://{host}:{port}"
            dist.init_process_group(
                rank=rank, world_size=world_size, backend=backend, init_method=init_method
            )
        self._register(ParallelMode.GLOBAL, rank, world_size, list(range(world_size)),
                       dist.group.WORLD)

    def init_parallel_groups(self):
        rank, world = self._rank, self._world_size
        args = (rank, world, self.tensor_parallel_size, self.pipeline_parallel_size,
                self.data_parallel_size, self.context_parallel_size)
        for init_cls in (
            TensorParallelGroupInitializer,
            PipelineParallelGroupInitializer,
            DataParallelGroupInitializer,
            ExpertDataParallelGroupInitializer,
            ContextParallelGroupInitializer,
        ):
            result = init_cls(*args).init_dist_group()
            assert result is not None
            self._register(result.parallel_mode, result.local_rank,
                           result.local_world_size, result.ranks_in_group,
                           result.process_group)

    def _register(self, mode, local_rank, local_world_size, ranks, group):
        self._groups[mode] = group
        self._ranks_in_group[mode] = ranks
        self._local_ranks[mode] = local_rank
        self._world_sizes[mode] = local_world_size

    def _set_context(self):
        global _GLOBAL_CONTEXT
        _GLOBAL_CONTEXT = self

    @staticmethod
    def get_context() -> Optional["ParallelContext"]:
        return _GLOBAL_CONTEXT

    def set_seed(self, seed: int):
        """Seed python/numpy/torch.  TP ranks get decorrelated CUDA seeds so
        dropout differs across tensor shards (Megatron convention)."""
        random.seed(seed)
        np.random.seed(seed)
        torch.manual_seed(seed)
        if torch.cuda.is_available():
            tp_seed = seed + 2718 * self.get_local_rank(ParallelMode.TENSOR)
            torch.cuda.manual_seed(tp_seed)

    # ---------------------------------------------------------------- queries

    def is_initialized(self, mode: ParallelMode) -> bool:
        return mode in self._groups

    def get_global_rank(self) -> int:
        return self._rank

    def get_local_rank(self, mode: ParallelMode = ParallelMode.GLOBAL) -> int:
        return self._local_ranks[mode]

    def get_world_size(self, mode: ParallelMode = ParallelMode.GLOBAL) -> int:
        return self._world_sizes[mode]

    def get_group(self, mode: ParallelMode):
        return self._groups[mode]

    def get_ranks_in_group(self, mode: ParallelMode) -> List[int]:
        return self._ranks_in_group[mode]

    def get_global_rank_from_local_rank(self, local_rank: int, mode: ParallelMode) -> int:
        return self._ranks_in_group[mode][local_rank]

    def is_first_rank(self, mode: ParallelMode) -> bool:
        return self.get_local_rank(mode) == 0

    def is_last_rank(self, mode: ParallelMode) -> bool:
        return self.get_local_rank(mode) == self.get_world_size(mode) - 1

    # pipeline neighbours -----------------------------------------------------

    def get_next_global_rank(self, mode: ParallelMode = ParallelMode.PIPELINE) -> int:
        ranks = self._ranks_in_group[mode]
        idx = self.get_local_rank(mode)
        return ranks[(idx + 1) % len(ranks)]

    def get_prev_global_rank(self, mode: ParallelMode = ParallelMode.PIPELINE) -> int:
        ranks = self._ranks_in_group[mode]
        idx = self.get_local_rank(mode)
        return ranks[(idx - 1) % len(ranks)]

    # device ------------------------------------------------------------------

    @property
    def device(self) -> torch.device:
        if torch.cuda.is_available():
            return torch.device("cuda", self._local_rank)
        return torch.device("cpu")

    def map_rank_to_device(self) -> dict:
        """global rank -> local device index (single-node: identity on local_rank)."""
        return {r: r % max(self._local_world_size, 1) for r in range(self._world_size)}

    # teardown ----------------------------------------------------------------

    def destroy(self):
        global _GLOBAL_CONTEXT
        if dist.is_initialized():
            dist.barrier()
            for mode, group in self._groups.items():
                if mode != ParallelMode.GLOBAL and group is not None:
                    dist.destroy_process_group(group)
            dist.destroy_process_group()
        self._groups.clear()
        _GLOBAL_CONTEXT = None
