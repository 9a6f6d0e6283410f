"""Model partitioning into pipeline stages.

Reference parity: nn/pipeline_parallel/partitioner.py used
transformers.utils.fx symbolic tracing + per-node param counting.  Here the
primary path is structural: transformer models (native pipegoose_amd models
AND HF models) expose {embedding+pre-layers, a block list, post-layers+head};
we cut the block list by parameter count (optionally by measured memory via
partitioning/profile.py) and wrap each stage as a plain nn.Sequential-like
module.  This avoids fx-trace brittleness on modern HF models (SURVEY 'hard
parts') while producing the same uniform cuts; an fx fallback can be added
per-architecture via `register_structure`.
"""
from typing import Callable, List, Optional

from torch import nn

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class ModelStructure:
    """A transformer decomposed into: pre (embeddings...), blocks, post (final
    norm / head).  `pre` runs on stage 0, `post` on the last stage."""

    def __init__(self, pre: List[nn.Module], blocks: List[nn.Module],
                 post: List[nn.Module]):
        self.pre = pre
        self.blocks = blocks
        self.post = post


_STRUCTURE_RESOLVERS: List[Callable[[nn.Module], Optional[ModelStructure]]] = []


def register_structure(resolver: Callable[[nn.Module], Optional[ModelStructure]]):
    _STRUCTURE_RESOLVERS.append(resolver)


def _resolve_native_bloom(model: nn.Module) -> Optional[ModelStructure]:
    from pipegoose_amd.models.bloom import BloomForCausalLM, BloomModel
    if isinstance(model, BloomForCausalLM):
        tr = model.transformer
        return ModelStructure(
            pre=[tr.word_embeddings, tr.word_embeddings_layernorm],
            blocks=list(tr.h),
            post=[tr.ln_f, model.lm_head],
        )
    if isinstance(model, BloomModel):
        return ModelStructure(
            pre=[model.word_embeddings, model.word_embeddings_layernorm],
            blocks=list(model.h),
            post=[model.ln_f],
        )
    return None


def _resolve_hf_transformer(model: nn.Module) -> Optional[ModelStructure]:
    """Generic HF causal-LM shape: model.<base>.{embeddings..., h|layers, ln_f}
    (+ lm_head)."""
    base = getattr(model, getattr(model, "base_model_prefix", ""), None) or model
    blocks_attr = None
    for name in ("h", "layers", "layer"):
        if hasattr(base, name) and isinstance(getattr(base, name), nn.ModuleList):
            blocks_attr = name
            break
    if blocks_attr is None:
        return None
    blocks = list(getattr(base, blocks_attr))
    pre, post = [], []
    seen_blocks = False
    for name, child in base.named_children():
        if name == blocks_attr:
            seen_blocks = True
            continue
        (post if seen_blocks else pre).append(child)
    if model is not base:
        for name, child in model.named_children():
            if child is not base:
                post.append(child)
    return ModelStructure(pre=pre, blocks=blocks, post=post)


def _resolve_native_llama(model: nn.Module) -> Optional[ModelStructure]:
    from pipegoose_amd.models.llama import LlamaForCausalLM, LlamaModel
    if isinstance(model, LlamaForCausalLM):
        m = model.model
        return ModelStructure(
            pre=[m.embed_tokens],
            blocks=list(m.layers),
            post=[m.norm, model.lm_head],
        )
    if isinstance(model, LlamaModel):
        return ModelStructure(pre=[model.embed_tokens], blocks=list(model.layers),
                              post=[model.norm])
    return None


def _resolve_sequential(model: nn.Module) -> Optional[ModelStructure]:
    if isinstance(model, nn.Sequential):
        return ModelStructure(pre=[], blocks=list(model), post=[])
    return None


_STRUCTURE_RESOLVERS.extend(
    [_resolve_sequential, _resolve_native_bloom, _resolve_native_llama,
     _resolve_hf_transformer])


def _param_bytes(m: nn.Module) -> int:
    return sum(p.numel() * p.element_size() for p in m.parameters())


class PartitionStage(nn.Module):
    """One pipeline stage: a plain module chain over the hidden state.

    Stage 0 consumes input_ids through `pre`; the last stage ends with `post`.
    """

    def __init__(self, modules: List[nn.Module]):
        super().__init__()
        self.chain = nn.ModuleList(modules)

    def forward(self, x):
        for m in self.chain:
            x = m(x)
        return x


class UniformPartitioner:
    """Cut the block list so per-stage parameter bytes are balanced.

    Like the reference (partitioner.py:77-93), embedding/head params are
    excluded from the balance target (they sit on first/last stage anyway);
    optionally pass `sizes` (e.g. from ProfileByMemory) to balance by measured
    activation+param memory instead.
    """

    def __init__(self, model: nn.Module, parallel_context: ParallelContext,
                 sizes: Optional[List[float]] = None):
        self.model = model
        self.parallel_context = parallel_context
        self.sizes = sizes

    def _structure(self) -> ModelStructure:
        for resolver in _STRUCTURE_RESOLVERS:
            s = resolver(self.model)
            if s is not None:
                return s
        raise ValueError(
            f"cannot determine pipeline structure of {type(self.model).__name__}; "
            "register one with pipegoose_amd.nn.pipeline_parallel.partitioner."
            "register_structure")

    def split(self, n_partitions: Optional[int] = None) -> List[PartitionStage]:
        pp = n_partitions or self.parallel_context.get_world_size(ParallelMode.PIPELINE)
        s = self._structure()
        n_blocks = len(s.blocks)
        assert n_blocks >= pp, f"{n_blocks} blocks < {pp} stages"

        weights = self.sizes or [_param_bytes(b) for b in s.blocks]
        total = sum(weights)
        # greedy contiguous split: cut when the running sum passes i/pp of total
        cuts = []
        acc = 0.0
        for i, w in enumerate(weights):
            acc += w
            if len(cuts) < pp - 1 and acc >= total * (len(cuts) + 1) / pp:
                cuts.append(i + 1)
        while len(cuts) < pp - 1:
            cuts.append(n_blocks)
        bounds = [0] + cuts + [n_blocks]

        stages = []
        for r in range(pp):
            mods: List[nn.Module] = []
            if r == 0:
                mods.extend(s.pre)
            mods.extend(s.blocks[bounds[r]:bounds[r + 1]])
            if r == pp - 1:
                mods.extend(s.post)
            stages.append(PartitionStage(mods))
        return stages

    def get_model_partition(self) -> PartitionStage:
        rank = self.parallel_context.get_local_rank(ParallelMode.PIPELINE)
        return self.split()[rank]
