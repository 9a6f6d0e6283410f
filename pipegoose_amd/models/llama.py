"""Native LLaMA-style model family, tensor-parallel by construction.

Second model family next to BLOOM (models/bloom.py): RMSNorm (fused HIP
kernel), rotary position embeddings (in-kernel cos/sin, ops/rope.py), SwiGLU
MLP, untied vocab-parallel LM head.  Attention reuses the hand-written flash
kernel with the ALiBi slopes set to zero (causal, no bias).

Module names follow the HF llama layout (model.layers.N.self_attn.q_proj,
mlp.gate_proj, ...) so TP mappings and checkpoints line up.
"""
import math
from dataclasses import dataclass

import torch
from torch import nn
import torch.nn.functional as TF

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.tensor_parallel.embedding import ParallelEmbedding
from pipegoose_amd.nn.tensor_parallel.linear import (ColumnParallelLinear,
                                                     RowParallelLinear)
from pipegoose_amd.nn.tensor_parallel.loss import VocabParallelCrossEntropy
from pipegoose_amd.nn.tensor_parallel.rms_norm import RMSNorm
from pipegoose_amd.ops.rope import apply_rope


@dataclass
class LlamaConfig:
    vocab_size: int = 32000
    hidden_size: int = 4096
    intermediate_size: int = 11008
    n_layer: int = 32
    n_head: int = 32
    n_kv_head: int = 0        # 0 = MHA; else GQA (q heads share kv heads)
    rope_theta: float = 10000.0
    rms_norm_eps: float = 1e-6
    initializer_range: float = 0.02
    sequence_parallel: bool = False
    # sequences shard along S over the CONTEXT group (ring attention);
    # params replicate over CP — see models/bloom.py BloomConfig
    context_parallel: bool = False

    @property
    def head_dim(self):
        return self.hidden_size // self.n_head


def llama_7b():
    return LlamaConfig()


def llama_1b():
    return LlamaConfig(hidden_size=2048, intermediate_size=5504,
                       n_layer=16, n_head=16)


def llama_tiny():
    """For tests."""
    return LlamaConfig(vocab_size=256, hidden_size=64, intermediate_size=128,
                       n_layer=2, n_head=4)


class LlamaAttention(nn.Module):
    def __init__(self, config: LlamaConfig, parallel_context: ParallelContext):
        super().__init__()
        tp = parallel_context.get_world_size(ParallelMode.TENSOR)
        assert config.n_head % tp == 0
        n_kv = config.n_kv_head or config.n_head
        assert config.n_head % n_kv == 0 and n_kv % tp == 0
        self.num_heads = config.n_head // tp
        self.num_kv_heads = n_kv // tp
        self.kv_group = self.num_heads // self.num_kv_heads
        self.head_dim = config.head_dim
        self.rope_theta = config.rope_theta
        self.inv_norm = 1.0 / math.sqrt(self.head_dim)
        sp = config.sequence_parallel
        h = config.hidden_size
        kv_out = n_kv * config.head_dim
        self.q_proj = ColumnParallelLinear(h, h, bias=False, sequence_parallel=sp,
                                           parallel_context=parallel_context)
        self.k_proj = ColumnParallelLinear(h, kv_out, bias=False,
                                           sequence_parallel=sp,
                                           parallel_context=parallel_context)
        self.v_proj = ColumnParallelLinear(h, kv_out, bias=False,
                                           sequence_parallel=sp,
                                           parallel_context=parallel_context)
        self.o_proj = RowParallelLinear(h, h, bias=False, sequence_parallel=sp,
                                        parallel_context=parallel_context)
        self.register_buffer("zero_slopes", torch.zeros(self.num_heads),
                             persistent=False)
        self._mask_cache = {}
        self.config = config
        self.parallel_context = parallel_context

    def _cp_size(self) -> int:
        if not getattr(self.config, "context_parallel", False):
            return 1
        return self.parallel_context.get_world_size(ParallelMode.CONTEXT)

    def _causal_mask(self, S: int, device, dtype):
        key = (S, device, dtype)
        if key not in self._mask_cache:
            m = torch.triu(torch.full((S, S), float("-inf"), device=device), 1)
            self._mask_cache[key] = m.to(dtype)[None, None]
        return self._mask_cache[key]

    def forward(self, hidden: torch.Tensor, past_kv=None,
                use_cache: bool = False):
        B = hidden.size(0)
        q = self.q_proj(hidden)
        S = q.size(1)
        k = self.k_proj(hidden)
        v = self.v_proj(hidden)
        q = q.view(B, S, self.num_heads, self.head_dim).transpose(1, 2)
        k = k.view(B, S, self.num_kv_heads, self.head_dim).transpose(1, 2)
        v = v.view(B, S, self.num_kv_heads, self.head_dim).transpose(1, 2)
        graph_mode = past_kv is not None and getattr(past_kv, "graph_mode",
                                                     False)
        if graph_mode:
            # hipGraph decode: the absolute position is the cache's device
            # pos_t — no host value may enter the captured step
            from pipegoose_amd.ops.rope import rope_at_position
            q = rope_at_position(q, self.rope_theta, past_kv.pos_t)
            k = rope_at_position(k, self.rope_theta, past_kv.pos_t)
        else:
            if past_kv is None:
                offset = 0
            elif hasattr(past_kv, "len"):
                offset = past_kv.len
            else:
                offset = past_kv[0].size(2)
            if self._cp_size() > 1 and past_kv is None:
                # context parallelism: positions are global, shards local
                offset += self.parallel_context.get_local_rank(
                    ParallelMode.CONTEXT) * S
            q = apply_rope(q, self.rope_theta, pos_offset=offset)
            k = apply_rope(k, self.rope_theta, pos_offset=offset)
        present = None
        if past_kv is not None:
            if hasattr(past_kv, "append"):     # StaticKVCache layer
                k, v = past_kv.append(k, v)
                present = past_kv
            else:
                k = torch.cat([past_kv[0], k], dim=2)
                v = torch.cat([past_kv[1], v], dim=2)
        if use_cache and present is None:
            present = (k, v)

        from pipegoose_amd.ops.attention import (_kernel_supported,
                                                 alibi_attention)
        kernel_ok = (_kernel_supported(q) and past_kv is None
                     and not use_cache)
        if self._cp_size() > 1 and past_kv is None and not use_cache:
            from pipegoose_amd.nn.ring_attention import ring_attention_rotate
            ke = k.repeat_interleave(self.kv_group, dim=1) \
                if self.kv_group > 1 else k
            ve = v.repeat_interleave(self.kv_group, dim=1) \
                if self.kv_group > 1 else v
            out = ring_attention_rotate(
                q.contiguous(), ke.contiguous(), ve.contiguous(),
                torch.zeros(self.num_heads, device=q.device),
                self.inv_norm, parallel_context=self.parallel_context,
                parallel_mode=ParallelMode.CONTEXT).to(q.dtype)
        elif graph_mode:
            # full-length masked attention over the static cache (constant
            # shapes; unfilled slots are zero + masked by -inf)
            ke = k.repeat_interleave(self.kv_group, dim=1) \
                if self.kv_group > 1 else k
            ve = v.repeat_interleave(self.kv_group, dim=1) \
                if self.kv_group > 1 else v
            kl = ke.size(2)
            if not hasattr(self, "_kpos_cache"):
                self._kpos_cache = {}
            kpos = self._kpos_cache.get((kl, q.device))
            if kpos is None:
                kpos = torch.arange(kl, device=q.device)
                self._kpos_cache[(kl, q.device)] = kpos
            bias = torch.zeros(1, 1, 1, kl, device=q.device, dtype=q.dtype)
            bias = bias.masked_fill((kpos > past_kv.pos_t)[None, None, None],
                                    float("-inf"))
            out = TF.scaled_dot_product_attention(
                q, ke, ve, attn_mask=bias, scale=self.inv_norm)
        elif k.size(2) != S or (self.kv_group > 1 and not kernel_ok):
            # decode / CPU-GQA fallback: expand kv heads, rect mask, sdpa
            ke = k.repeat_interleave(self.kv_group, dim=1) \
                if self.kv_group > 1 else k
            ve = v.repeat_interleave(self.kv_group, dim=1) \
                if self.kv_group > 1 else v
            kl = ke.size(2)
            rel = torch.arange(kl, device=q.device)[None, :] \
                - torch.arange(kl - S, kl, device=q.device)[:, None]
            bias = torch.zeros(S, kl, device=q.device, dtype=q.dtype)
            bias = bias.masked_fill(rel > 0, float("-inf"))[None, None]
            out = TF.scaled_dot_product_attention(
                q, ke, ve, attn_mask=bias, scale=self.inv_norm)
        else:
            # the hand-written kernel handles GQA natively (kv head =
            # q head / group — attention.hip)
            out = alibi_attention(
                q, k, v, self.zero_slopes, self.inv_norm,
                mask_fallback=lambda s, dev, dt: self._causal_mask(s, dev, dt))
        out = out.transpose(1, 2).reshape(B, S, self.num_heads * self.head_dim)
        out = self.o_proj(out)
        return (out, present) if use_cache else out


class LlamaMLP(nn.Module):
    def __init__(self, config: LlamaConfig, parallel_context: ParallelContext):
        super().__init__()
        h, inter = config.hidden_size, config.intermediate_size
        sp = config.sequence_parallel
        self.gate_proj = ColumnParallelLinear(h, inter, bias=False,
                                              sequence_parallel=sp,
                                              parallel_context=parallel_context)
        self.up_proj = ColumnParallelLinear(h, inter, bias=False,
                                            sequence_parallel=sp,
                                            parallel_context=parallel_context)
        self.down_proj = RowParallelLinear(inter, h, bias=False,
                                           sequence_parallel=sp,
                                           parallel_context=parallel_context)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        from pipegoose_amd.ops.silu_mul import silu_mul
        # fused silu(gate)*up: one HBM pass (ops/csrc/silu_mul.hip)
        return self.down_proj(silu_mul(self.gate_proj(hidden),
                                       self.up_proj(hidden)))


class LlamaBlock(nn.Module):
    def __init__(self, config: LlamaConfig, parallel_context: ParallelContext):
        super().__init__()
        sp = config.sequence_parallel
        ctx = parallel_context if sp else None
        self.input_layernorm = RMSNorm(config.hidden_size, config.rms_norm_eps,
                                       sequence_parallel=sp, parallel_context=ctx)
        self.self_attn = LlamaAttention(config, parallel_context)
        self.post_attention_layernorm = RMSNorm(
            config.hidden_size, config.rms_norm_eps,
            sequence_parallel=sp, parallel_context=ctx)
        self.mlp = LlamaMLP(config, parallel_context)

    def forward(self, hidden: torch.Tensor, past_kv=None,
                use_cache: bool = False):
        attn_out = self.self_attn(self.input_layernorm(hidden),
                                  past_kv=past_kv, use_cache=use_cache)
        present = None
        if use_cache:
            attn_out, present = attn_out
        hidden = hidden + attn_out
        hidden = hidden + self.mlp(self.post_attention_layernorm(hidden))
        return (hidden, present) if use_cache else hidden


class LlamaModel(nn.Module):
    def __init__(self, config: LlamaConfig, parallel_context: ParallelContext):
        super().__init__()
        self.config = config
        self.embed_tokens = ParallelEmbedding(
            config.vocab_size, config.hidden_size,
            sequence_parallel=config.sequence_parallel,
            parallel_context=parallel_context)
        self.layers = nn.ModuleList(
            [LlamaBlock(config, parallel_context) for _ in range(config.n_layer)])
        self.norm = RMSNorm(
            config.hidden_size, config.rms_norm_eps,
            sequence_parallel=config.sequence_parallel,
            parallel_context=parallel_context if config.sequence_parallel else None)

    def forward(self, input_ids: torch.Tensor, past=None,
                use_cache: bool = False):
        hidden = self.embed_tokens(input_ids)
        if use_cache or past is not None:
            presents = []
            for i, layer in enumerate(self.layers):
                pk = past[i] if past is not None else None
                hidden, present = layer(hidden, past_kv=pk, use_cache=True)
                presents.append(present)
            return self.norm(hidden), presents
        if getattr(self, "gradient_checkpointing", False) and self.training:
            import torch.utils.checkpoint as ckpt
            for layer in self.layers:
                hidden = ckpt.checkpoint(layer, hidden, use_reentrant=False)
        else:
            for layer in self.layers:
                hidden = layer(hidden)
        return self.norm(hidden)


class LlamaForCausalLM(nn.Module):
    def __init__(self, config: LlamaConfig, parallel_context: ParallelContext):
        super().__init__()
        self.config = config
        self.parallel_context = parallel_context
        self.model = LlamaModel(config, parallel_context)
        tp = parallel_context.get_world_size(ParallelMode.TENSOR)
        self.lm_head = ColumnParallelLinear(
            config.hidden_size, config.vocab_size, bias=False,
            gather_output=False, sequence_parallel=config.sequence_parallel,
            parallel_context=parallel_context)
        self.loss_fn = VocabParallelCrossEntropy(parallel_context=parallel_context) \
            if tp > 1 else None
        self.apply(self._init_weights)

    def _init_weights(self, module):
        std = self.config.initializer_range
        if isinstance(module, (ColumnParallelLinear, RowParallelLinear)):
            nn.init.normal_(module.weight, mean=0.0, std=std)
            if module.bias is not None:
                nn.init.zeros_(module.bias)
        elif isinstance(module, ParallelEmbedding):
            nn.init.normal_(module.weight, mean=0.0, std=std)

    def forward(self, input_ids: torch.Tensor, labels: torch.Tensor = None,
                past=None, use_cache: bool = False):
        if use_cache or past is not None:
            hidden, presents = self.model(input_ids, past=past, use_cache=True)
            return self.lm_head(hidden), presents
        hidden = self.model(input_ids)
        logits = self.lm_head(hidden)
        if labels is None:
            return logits
        from pipegoose_amd.models.bloom import _shifted_labels
        shift_labels = _shifted_labels(labels)
        if self.loss_fn is not None:
            return self.loss_fn(logits, shift_labels)
        if logits.is_cuda:
            from pipegoose_amd.ops.cross_entropy import fused_cross_entropy
            return fused_cross_entropy(
                logits.reshape(-1, logits.size(-1)), shift_labels.reshape(-1))
        return TF.cross_entropy(
            logits.float().reshape(-1, logits.size(-1)),
            shift_labels.reshape(-1), ignore_index=-100)


    def new_kv_cache(self, batch_size: int, max_len: int):
        """Preallocated static KV cache for decode (models/kv_cache.py)."""
        from pipegoose_amd.models.kv_cache import StaticKVCache
        attn = self.model.layers[0].self_attn
        p = next(self.parameters())
        return StaticKVCache(len(self.model.layers), batch_size,
                             attn.num_kv_heads, max_len, attn.head_dim,
                             p.dtype, p.device)

    def new_graph_kv_cache(self, batch_size: int, max_len: int):
        """Cache bank for hipGraph-captured decode (models/graph_decode.py)."""
        from pipegoose_amd.models.kv_cache import GraphKVCache
        attn = self.model.layers[0].self_attn
        p = next(self.parameters())
        return GraphKVCache(len(self.model.layers), batch_size,
                            attn.num_kv_heads, max_len, attn.head_dim,
                            p.dtype, p.device)

    def gradient_checkpointing_enable(self, enabled: bool = True):
        """Recompute each block in backward instead of storing activations —
        trades ~30% step time for O(sqrt) activation memory (capability the
        reference lacked; composes with TP/SP/DP)."""
        self.model.gradient_checkpointing = enabled

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 20,
                 temperature: float = 0.0, top_k: int = 0,
                 eos_token_id=None):
        from pipegoose_amd.models.generation import generate as _generate
        return _generate(self, input_ids, max_new_tokens=max_new_tokens,
                         temperature=temperature, top_k=top_k,
                         parallel_context=self.parallel_context,
                         eos_token_id=eos_token_id)
