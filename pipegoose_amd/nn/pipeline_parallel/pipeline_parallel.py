"""PipelineParallel wrapper: partition the model, keep this rank's stage,
expose a module whose forward runs the 1F1B engine.

Reference parity: nn/pipeline_parallel/pipeline_parallel.py:26-50 (same
user-facing shape: wrap, parallelize, call the model).  The returned module's
``forward(inputs, labels=None)`` executes one full pipelined step (forward
+ backward when labels given) and returns the mean loss on the last stage.
"""
from typing import Callable, Optional

from torch import nn

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.parallel import Parallel
from pipegoose_amd.nn.pipeline_parallel.engine import PipelineEngine
from pipegoose_amd.nn.pipeline_parallel.partitioner import UniformPartitioner


class PipelineStageModule(nn.Module):
    def __init__(self, engine: PipelineEngine):
        super().__init__()
        self.stage = engine.stage  # registers stage params
        self.engine = engine

    def forward(self, inputs=None, labels=None, input_ids=None):
        # accept both positional and Trainer-style input_ids= kwargs
        if inputs is None:
            inputs = input_ids
        # DataParallel wraps THIS module; hand its hook-controller to the
        # engine so grad sync is deferred to the end of the microbatch loop
        return self.engine.run(inputs, labels,
                               dp=getattr(self, "_dp_wrapper", None))


class InterleavedStageModule(nn.Module):
    """Holds this rank's v model chunks; forward runs the interleaved
    1F1B engine (interleaved.py)."""

    def __init__(self, engine):
        super().__init__()
        self.chunks = nn.ModuleList(engine.chunks)  # registers params
        self.engine = engine

    def forward(self, inputs=None, labels=None, input_ids=None):
        if inputs is None:
            inputs = input_ids
        return self.engine.run(inputs, labels,
                               dp=getattr(self, "_dp_wrapper", None))


class PipelineParallel(Parallel):
    def __init__(
        self,
        module: nn.Module,
        parallel_context: ParallelContext,
        n_microbatches: int = 1,
        schedule: str = "1f1b",
        loss_fn: Optional[Callable] = None,
        partition_sizes=None,
        moe_aux_weight: float = 0.01,
        moe_z_weight: float = 0.1,
        virtual_stages: int = 1,
    ):
        super().__init__(module, parallel_context)
        self.n_microbatches = n_microbatches
        self.schedule = schedule
        self.loss_fn = loss_fn
        self.partition_sizes = partition_sizes
        # Under PP, MoE aux/z losses are consumed by the ENGINE per stage and
        # per microbatch (ExpertLoss can't see non-last stages' routers):
        self.moe_aux_weight = moe_aux_weight
        self.moe_z_weight = moe_z_weight
        self.virtual_stages = virtual_stages
        if schedule == "interleaved":
            assert virtual_stages > 1, \
                "schedule='interleaved' needs virtual_stages > 1"

    def parallelize(self) -> nn.Module:
        pp = self.parallel_context.get_world_size(ParallelMode.PIPELINE)
        if pp == 1:
            return self.module
        self._untie_shared_weights()
        if self.schedule == "interleaved":
            from pipegoose_amd.nn.pipeline_parallel.interleaved import (
                InterleavedPipelineEngine)
            v = self.virtual_stages
            rank = self.parallel_context.get_local_rank(ParallelMode.PIPELINE)
            stages = UniformPartitioner(
                self.module, self.parallel_context,
                sizes=self.partition_sizes).split(pp * v)
            chunks = [stages[c * pp + rank] for c in range(v)]
            engine = InterleavedPipelineEngine(
                chunks, self.parallel_context, self.n_microbatches,
                loss_fn=self.loss_fn, moe_aux_weight=self.moe_aux_weight,
                moe_z_weight=self.moe_z_weight)
            wrapped = InterleavedStageModule(engine)
            self._save_metadata(wrapped, self.parallel_context)
            return wrapped
        stage = UniformPartitioner(
            self.module, self.parallel_context, sizes=self.partition_sizes
        ).get_model_partition()
        engine = PipelineEngine(
            stage, self.parallel_context, self.n_microbatches,
            schedule=self.schedule, loss_fn=self.loss_fn,
            moe_aux_weight=self.moe_aux_weight,
            moe_z_weight=self.moe_z_weight)
        wrapped = PipelineStageModule(engine)
        self._save_metadata(wrapped, self.parallel_context)
        return wrapped

    def _untie_shared_weights(self):
        """Embedding/LM-head weight tying cannot span pipeline stages; clone
        the tied weight so each stage owns its parameter (forward-identical;
        Megatron-style tied-grad all-reduce is a later optimization)."""
        from pipegoose_amd.nn.tensor_parallel.embedding import ParallelEmbedding
        emb_weights = {id(m.weight) for m in self.module.modules()
                       if isinstance(m, (nn.Embedding, ParallelEmbedding))}
        for m in self.module.modules():
            if isinstance(m, nn.Embedding) or isinstance(m, ParallelEmbedding):
                continue
            w = getattr(m, "weight", None)
            if w is not None and id(w) in emb_weights:
                m.weight = nn.Parameter(w.detach().clone())
