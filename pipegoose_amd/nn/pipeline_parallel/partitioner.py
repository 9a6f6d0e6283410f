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


class _FxStage(nn.Module):
    """One fx-derived pipeline stage.

    Receives the tuple of live values crossing its input boundary, runs its
    split_module submodule, and emits the tuple of values live at its output
    boundary (which may pass earlier values through untouched — skip
    connections spanning stages).  Single-value boundaries collapse to a bare
    tensor so the engine's single-tensor fast path applies.
    """

    def __init__(self, submod: nn.Module, in_names, arg_specs, out_specs):
        super().__init__()
        self.submod = submod
        self.in_names = list(in_names)      # wire-in value names, in order
        self.arg_specs = list(arg_specs)    # ("n", name) | ("c", literal)
        self.out_specs = list(out_specs)    # (name, src): "wire"|"whole"|idx

    def forward(self, *wire_vals):
        env = dict(zip(self.in_names, wire_vals))
        res = self.submod(*[env[a] if tag == "n" else a
                            for tag, a in self.arg_specs])
        outs = []
        for name, src in self.out_specs:
            if src == "wire":
                outs.append(env[name])
            elif src == "whole":
                outs.append(res)
            else:  # tuple element index of the submod result
                outs.append(res[src])
        return outs[0] if len(outs) == 1 else tuple(outs)


class FxUniformPartitioner:
    """torch.fx fallback partitioner with cross-shard value stitching.

    For models the structural resolvers reject (arbitrary inter-block
    dataflow): symbolically trace, assign graph nodes to `pp` contiguous
    shards balanced by parameter bytes, split with
    torch.fx.passes.split_module (which threads every cross-shard value),
    then wrap each shard so the live value set travels stage-to-stage as a
    tuple (the engine's boundary type).  Subsumes the reference's
    fx partitioner (pipegoose/nn/pipeline_parallel/partitioner.py:55-219)
    including its cross-shard value propagation, without its
    transformer-block-boundary restriction.

    `trace_fn(model) -> GraphModule` may be supplied (e.g. a
    transformers.utils.fx tracer pinned to input_ids); default is plain
    torch.fx.symbolic_trace.
    """

    def __init__(self, model: nn.Module, parallel_context: ParallelContext,
                 trace_fn: Optional[Callable] = None):
        self.model = model
        self.parallel_context = parallel_context
        self.trace_fn = trace_fn

    def _trace(self):
        import torch.fx
        if self.trace_fn is not None:
            return self.trace_fn(self.model)
        return torch.fx.symbolic_trace(self.model)

    def split(self, n_partitions: Optional[int] = None):
        import torch.fx
        from torch.fx.passes.split_module import split_module

        pp = n_partitions or self.parallel_context.get_world_size(
            ParallelMode.PIPELINE)
        gm = self._trace()

        # parameter bytes per node (call_module leaves + get_attr params)
        def node_bytes(node):
            if node.op == "call_module":
                sub = gm.get_submodule(node.target)
                return sum(p.numel() * p.element_size()
                           for p in sub.parameters())
            if node.op == "get_attr":
                try:
                    t = gm.get_parameter(node.target)
                    return t.numel() * t.element_size()
                except AttributeError:
                    return 0
            return 0

        nodes = [n for n in gm.graph.nodes if n.op not in ("placeholder",
                                                           "output")]
        weights = [node_bytes(n) for n in nodes]
        total = max(sum(weights), 1)
        part_of = {}
        acc, part = 0.0, 0
        for n, w in zip(nodes, weights):
            part_of[n.name] = part
            acc += w
            if part < pp - 1 and acc >= total * (part + 1) / pp:
                part += 1
        for n in gm.graph.nodes:
            if n.op == "placeholder":
                part_of[n.name] = 0

        splitted = split_module(gm, self.model,
                                lambda node: part_of.get(node.name, pp - 1))

        # Walk the top-level stitched graph to derive per-stage wiring.
        # env value names = top-graph node names; a stage consumes its
        # submod's args, plus passes through values still needed later.
        g = splitted.graph
        placeholders = [n for n in g.nodes if n.op == "placeholder"]
        submod_calls = []  # (stage_idx, node)
        for n in g.nodes:
            if n.op == "call_module" and n.target.startswith("submod_"):
                submod_calls.append((int(n.target.split("_")[1]), n))
        submod_calls.sort(key=lambda t: t[0])
        n_stages = len(submod_calls)
        assert n_stages == pp, \
            f"fx split produced {n_stages} stages (wanted {pp}); " \
            "model too small for this pp degree"

        # producer stage per top-graph value (-1 = model input); getitem
        # nodes extracting a tuple element of a submod call belong to the
        # producing stage
        producer = {p.name: -1 for p in placeholders}
        getitems = {}  # value name -> (submod call name, index)
        for k, (_, call) in enumerate(submod_calls):
            producer[call.name] = k
        for n in g.nodes:
            if (n.op == "call_function"
                    and getattr(n.target, "__name__", "") == "getitem"
                    and getattr(n.args[0], "op", None) == "call_module"):
                producer[n.name] = producer[n.args[0].name]
                getitems[n.name] = (n.args[0].name, n.args[1])

        # last stage a value is needed by (output counts as the last stage)
        need_until = {}
        for k, (_, call) in enumerate(submod_calls):
            for a in call.args:
                if hasattr(a, "name"):
                    need_until[a.name] = max(need_until.get(a.name, -1), k)
        out_node = next(n for n in g.nodes if n.op == "output")
        out_args = out_node.args[0]
        out_args = list(out_args) if isinstance(out_args, (tuple, list)) \
            else [out_args]
        for a in out_args:
            if hasattr(a, "name"):
                need_until[a.name] = n_stages - 1

        order = {n.name: i for i, n in enumerate(g.nodes)}

        def spec_for(name, k, call):
            """How stage k materializes `name` for its output tuple."""
            if producer[name] < k:
                return (name, "wire")
            if name == call.name:
                return (name, "whole")
            assert name in getitems and getitems[name][0] == call.name, name
            return (name, getitems[name][1])

        stages = []
        for k, (_, call) in enumerate(submod_calls):
            if k == 0:
                in_names = [p.name for p in placeholders]
            else:
                in_names = sorted(
                    [nm for nm, p in producer.items()
                     if p < k and need_until.get(nm, -1) >= k],
                    key=lambda nm: order[nm])
            arg_specs = [("n", a.name) if hasattr(a, "name") else ("c", a)
                         for a in call.args]
            if k < n_stages - 1:
                out_specs = [spec_for(nm, k, call)
                             for nm, p in sorted(producer.items(),
                                                 key=lambda kv: order[kv[0]])
                             if p <= k and need_until.get(nm, -1) > k]
            else:
                out_specs = [spec_for(a.name, k, call) for a in out_args
                             if hasattr(a, "name")]
            stages.append(_FxStage(getattr(splitted, call.target),
                                   in_names, arg_specs, out_specs))
        return stages

    def get_model_partition(self):
        rank = self.parallel_context.get_local_rank(ParallelMode.PIPELINE)
        return self.split()[rank]
