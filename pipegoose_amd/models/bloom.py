"""Native BLOOM model family, tensor-parallel by construction.

This is the framework's flagship model (BASELINE.json configs: bloom-560m,
bloom-1b7, bloom-7b1): ALiBi attention, LayerNorm, GELU MLP, tied LM head —
behaviorally the BLOOM architecture, implemented MI355X-first:

  - built directly on Column/RowParallelLinear + ParallelEmbedding, so TP
    needs no module surgery and head counts are local by construction;
  - LayerNorm runs the fused CDNA4 HIP kernel (ops/layer_norm);
  - MLP uses the fused bias-GeLU kernel on GPU (ops/fused_bias_gelu);
  - loss is the vocab-parallel streaming CE (no full-logit gather when tp>1).

Module names match TensorParallelMapping ("self_attention.query_key_value",
"mlp.dense_h_to_4h", ...) so checkpoints line up with HF-surgered models.
"""
import math
from dataclasses import dataclass

import torch
from torch import nn
import torch.nn.functional as TF

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.tensor_parallel.embedding import ParallelEmbedding
from pipegoose_amd.nn.tensor_parallel.layer_norm import LayerNorm
from pipegoose_amd.nn.tensor_parallel.linear import ColumnParallelLinear, RowParallelLinear
from pipegoose_amd.nn.tensor_parallel.loss import VocabParallelCrossEntropy
from pipegoose_amd.ops.fused_bias_gelu import fused_bias_gelu


@dataclass
class BloomConfig:
    vocab_size: int = 250880
    hidden_size: int = 1024
    n_layer: int = 24
    n_head: int = 16
    layer_norm_epsilon: float = 1e-5
    initializer_range: float = 0.02
    # Megatron-style sequence parallelism (BASELINE config 5): LayerNorms and
    # residuals run on [B, S/tp, H] shards; TP boundaries become
    # all-gather / reduce-scatter along S.  Requires S % tp == 0.
    sequence_parallel: bool = False
    # Context parallelism: sequences shard along S over the CONTEXT group;
    # attention runs as rotation-based ring attention (nn/ring_attention.py).
    # Requires S % cp == 0; params are replicated over CP (sync grads with
    # DataParallel(mode=ParallelMode.CONTEXT)).
    context_parallel: bool = False

    @property
    def head_dim(self):
        return self.hidden_size // self.n_head


def bloom_560m():
    return BloomConfig(hidden_size=1024, n_layer=24, n_head=16)


def bloom_1b7():
    return BloomConfig(hidden_size=2048, n_layer=24, n_head=16)


def bloom_7b1():
    return BloomConfig(hidden_size=4096, n_layer=30, n_head=32)


def bloom_tiny():
    """For tests."""
    return BloomConfig(vocab_size=256, hidden_size=64, n_layer=2, n_head=4)


def alibi_slopes(n_head: int) -> torch.Tensor:
    """ALiBi per-head slopes (standard construction, handles non-pow2 heads)."""
    def pow2_slopes(n):
        start = 2.0 ** (-(2.0 ** -(math.log2(n) - 3)))
        return [start * (start ** i) for i in range(n)]

    if math.log2(n_head).is_integer():
        return torch.tensor(pow2_slopes(n_head))
    closest = 2 ** math.floor(math.log2(n_head))
    slopes = pow2_slopes(closest)
    extra = pow2_slopes(2 * closest)[0::2][: n_head - closest]
    return torch.tensor(slopes + extra)


class BloomAttention(nn.Module):
    def __init__(self, config: BloomConfig, parallel_context: ParallelContext):
        super().__init__()
        tp = parallel_context.get_world_size(ParallelMode.TENSOR)
        tp_rank = parallel_context.get_local_rank(ParallelMode.TENSOR)
        assert config.n_head % tp == 0
        self.num_heads = config.n_head // tp
        self.head_dim = config.head_dim
        self.hidden_size = config.hidden_size
        self.inv_norm = 1.0 / math.sqrt(self.head_dim)

        # QKV fused, grouped per head ([head][q|k|v][head_dim]) so the column
        # slice hands whole heads to each rank.
        sp = config.sequence_parallel
        self.query_key_value = ColumnParallelLinear(
            config.hidden_size, 3 * config.hidden_size,
            sequence_parallel=sp, parallel_context=parallel_context)
        self.dense = RowParallelLinear(
            config.hidden_size, config.hidden_size,
            sequence_parallel=sp, parallel_context=parallel_context)

        slopes = alibi_slopes(config.n_head)
        local = slopes.chunk(tp)[tp_rank].clone()
        self.register_buffer("alibi_slopes", local, persistent=False)
        self._bias_cache = {}
        self.config = config
        self.parallel_context = parallel_context

    def _alibi_bias(self, seq_len: int, device, dtype) -> torch.Tensor:
        key = (seq_len, device, dtype)
        cached = self._bias_cache.get(key)
        if cached is None:
            pos = torch.arange(seq_len, device=device)
            # bias[h, i, j] = slope_h * (j - i) for j <= i; -inf above diagonal
            rel = pos[None, :] - pos[:, None]
            bias = self.alibi_slopes.to(device=device, dtype=torch.float32)[:, None, None] \
                * rel[None, :, :].float()
            causal = torch.full((seq_len, seq_len), float("-inf"), device=device)
            causal = torch.triu(causal, diagonal=1)
            bias = (bias + causal[None]).to(dtype)
            cached = bias.unsqueeze(0)  # [1, H_local, S, S]
            self._bias_cache = {key: cached}
        return cached

    def _alibi_bias_rect(self, q_len: int, k_len: int, device, dtype):
        """Decode-path bias [1, H, q_len, k_len]: query rows sit at absolute
        positions k_len-q_len .. k_len-1."""
        qpos = torch.arange(k_len - q_len, k_len, device=device)
        kpos = torch.arange(k_len, device=device)
        rel = kpos[None, :] - qpos[:, None]                 # j - i_abs
        bias = self.alibi_slopes.to(device=device, dtype=torch.float32)[
            :, None, None] * rel[None].float()
        bias = bias.masked_fill(rel[None] > 0, float("-inf"))
        return bias.to(dtype)[None]

    def _alibi_bias_graph(self, k_len: int, pos_t: torch.Tensor, dtype):
        """Graph-mode decode bias [1, H, 1, k_len]: the single query sits at
        absolute position ``pos_t`` (a device tensor — no host value enters
        the graph); keys beyond it (unfilled cache slots) get -inf."""
        if not hasattr(self, "_kpos_cache"):
            self._kpos_cache = {}
        key = (k_len, pos_t.device)
        kpos = self._kpos_cache.get(key)
        if kpos is None:
            kpos = torch.arange(k_len, device=pos_t.device)
            self._kpos_cache[key] = kpos
        rel = (kpos - pos_t).float()                         # [k_len]
        bias = self.alibi_slopes.to(device=pos_t.device,
                                    dtype=torch.float32)[:, None, None] * rel
        bias = bias.masked_fill(kpos > pos_t, float("-inf"))  # [H, 1, k_len]
        return bias.to(dtype)[None]

    def forward(self, hidden: torch.Tensor, past_kv=None, use_cache: bool = False):
        B = hidden.size(0)
        fused = self.query_key_value(hidden)  # [B, S, local_heads * 3 * hd]
        S = fused.size(1)  # full sequence (SP mode all-gathered inside qkv)
        fused = fused.view(B, S, self.num_heads, 3, self.head_dim)
        q = fused[..., 0, :].transpose(1, 2)  # [B, H, S, hd]
        k = fused[..., 1, :].transpose(1, 2)
        v = fused[..., 2, :].transpose(1, 2)

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
                                                 alibi_attention,
                                                 alibi_attention_qkv)
        if present is not None and getattr(present, "graph_mode", False):
            # hipGraph-capturable decode: constant shapes (full cache
            # length), position enters as DATA via the cache's device pos_t
            bias = self._alibi_bias_graph(k.size(2), present.pos_t, q.dtype)
            out = TF.scaled_dot_product_attention(
                q, k, v, attn_mask=bias, scale=self.inv_norm)
        elif k.size(2) != S:
            # incremental decode: rectangular mask, torch sdpa
            bias = self._alibi_bias_rect(S, k.size(2), q.device, q.dtype)
            out = TF.scaled_dot_product_attention(
                q, k, v, attn_mask=bias, scale=self.inv_norm)
        elif (getattr(self.config, "context_parallel", False)
              and self.parallel_context.get_world_size(ParallelMode.CONTEXT) > 1
              and past_kv is None and not use_cache):
            # context parallelism: hidden is this rank's S-shard; ring
            # attention rotates KV around the CONTEXT group (ALiBi offsets
            # ride the blockwise kv_off)
            from pipegoose_amd.nn.ring_attention import ring_attention_rotate
            out = ring_attention_rotate(
                q.contiguous(), k.contiguous(), v.contiguous(),
                self.alibi_slopes.to(q.device, torch.float32), self.inv_norm,
                parallel_context=self.parallel_context,
                parallel_mode=ParallelMode.CONTEXT).to(q.dtype)
        elif past_kv is None and not use_cache and _kernel_supported(q):
            # training fast path: backward writes one d(fused) buffer
            out = alibi_attention_qkv(fused, self.alibi_slopes, self.inv_norm)
        else:
            # cached prefill / plain path: kernel when shapes allow, else sdpa
            out = alibi_attention(q, k, v, self.alibi_slopes, self.inv_norm,
                                  mask_fallback=self._alibi_bias)
        out = out.transpose(1, 2).reshape(B, S, self.num_heads * self.head_dim)
        out = self.dense(out)
        return (out, present) if use_cache else out


class BloomMLP(nn.Module):
    def __init__(self, config: BloomConfig, parallel_context: ParallelContext):
        super().__init__()
        h = config.hidden_size
        sp = config.sequence_parallel
        self.dense_h_to_4h = ColumnParallelLinear(
            h, 4 * h, sequence_parallel=sp, parallel_context=parallel_context)
        self.dense_4h_to_h = RowParallelLinear(
            4 * h, h, sequence_parallel=sp, parallel_context=parallel_context)

    def forward(self, hidden: torch.Tensor) -> torch.Tensor:
        # fused GEMM + bias + GeLU epilogue on GPU (ops/fused_bias_gelu)
        x = fused_bias_gelu(self.dense_h_to_4h, hidden)
        return self.dense_4h_to_h(x)


class BloomBlock(nn.Module):
    def __init__(self, config: BloomConfig, parallel_context: ParallelContext):
        super().__init__()
        eps = config.layer_norm_epsilon
        sp = config.sequence_parallel
        self.input_layernorm = LayerNorm(
            config.hidden_size, eps=eps, sequence_parallel=sp,
            parallel_context=parallel_context if sp else None)
        self.self_attention = BloomAttention(config, parallel_context)
        self.post_attention_layernorm = LayerNorm(
            config.hidden_size, eps=eps, sequence_parallel=sp,
            parallel_context=parallel_context if sp else None)
        self.mlp = BloomMLP(config, parallel_context)

    def forward(self, hidden: torch.Tensor, past_kv=None,
                use_cache: bool = False):
        attn_out = self.self_attention(self.input_layernorm(hidden),
                                       past_kv=past_kv, use_cache=use_cache)
        present = None
        if use_cache:
            attn_out, present = attn_out
        # residual add fused into the post-attention norm's HBM pass
        normed, hidden = self.post_attention_layernorm.forward_with_residual(
            attn_out, hidden)
        out = hidden + self.mlp(normed)
        return (out, present) if use_cache else out

    def forward_chained(self, hidden: torch.Tensor, pending):
        """Residual-chained form: the PREVIOUS block's mlp output arrives as
        ``pending`` and its residual add fuses into this block's input norm
        (BloomModel.forward drives this; plain ``forward`` keeps the
        single-tensor contract for pipeline stages)."""
        if pending is not None:
            normed, hidden = self.input_layernorm.forward_with_residual(
                pending, hidden)
        else:
            normed = self.input_layernorm(hidden)
        attn_out = self.self_attention(normed)
        normed2, hidden = self.post_attention_layernorm.forward_with_residual(
            attn_out, hidden)
        return hidden, self.mlp(normed2)


class BloomModel(nn.Module):
    def __init__(self, config: BloomConfig, parallel_context: ParallelContext):
        super().__init__()
        self.config = config
        eps = config.layer_norm_epsilon
        self.word_embeddings = ParallelEmbedding(
            config.vocab_size, config.hidden_size,
            sequence_parallel=config.sequence_parallel,
            parallel_context=parallel_context)
        sp = config.sequence_parallel
        self.word_embeddings_layernorm = LayerNorm(
            config.hidden_size, eps=eps, sequence_parallel=sp,
            parallel_context=parallel_context if sp else None)
        self.h = nn.ModuleList(
            [BloomBlock(config, parallel_context) for _ in range(config.n_layer)]
        )
        self.ln_f = LayerNorm(
            config.hidden_size, eps=eps, sequence_parallel=sp,
            parallel_context=parallel_context if sp else None)

    def forward(self, input_ids: torch.Tensor, past=None,
                use_cache: bool = False):
        hidden = self.word_embeddings_layernorm(self.word_embeddings(input_ids))
        if use_cache or past is not None:
            presents = []
            for i, block in enumerate(self.h):
                pk = past[i] if past is not None else None
                hidden, present = block(hidden, past_kv=pk, use_cache=True)
                presents.append(present)
            return self.ln_f(hidden), presents
        # training fast path — residual-chained blocks: every residual add
        # fuses into a norm's HBM pass (2 per block + the final one into ln_f)
        pending = None
        if getattr(self, "gradient_checkpointing", False) and self.training:
            import torch.utils.checkpoint as ckpt
            for block in self.h:
                hidden, pending = ckpt.checkpoint(
                    block.forward_chained, hidden, pending, use_reentrant=False)
        else:
            for block in self.h:
                hidden, pending = block.forward_chained(hidden, pending)
        normed, _ = self.ln_f.forward_with_residual(pending, hidden) \
            if pending is not None else (self.ln_f(hidden), hidden)
        return normed




def _shifted_labels(labels: torch.Tensor) -> torch.Tensor:
    """labels shifted left with a -100 pad at the end: CE over the FULL logits
    equals shift-slice CE but skips materializing logits[:, :-1].contiguous()
    (8.2 GB/step at BLOOM vocab) and its backward zeros+copy."""
    pad = labels.new_full((labels.size(0), 1), -100)
    return torch.cat([labels[:, 1:], pad], dim=1)

def make_causal_lm_loss(parallel_context: ParallelContext):
    """Shift-CE loss fn over (possibly vocab-sharded) logits — for the
    pipeline engine's last stage."""
    tp = parallel_context.get_world_size(ParallelMode.TENSOR)
    vp_ce = VocabParallelCrossEntropy(parallel_context=parallel_context) \
        if tp > 1 else None

    def loss_fn(logits, labels):
        shift_labels = _shifted_labels(labels)
        if vp_ce is not None:
            return vp_ce(logits, shift_labels)
        if logits.is_cuda:
            from pipegoose_amd.ops.cross_entropy import fused_cross_entropy
            return fused_cross_entropy(
                logits.reshape(-1, logits.size(-1)), shift_labels.reshape(-1))
        return TF.cross_entropy(
            logits.float().reshape(-1, logits.size(-1)),
            shift_labels.reshape(-1), ignore_index=-100)

    return loss_fn


class BloomForCausalLM(nn.Module):
    def __init__(self, config: BloomConfig, parallel_context: ParallelContext):
        super().__init__()
        self.config = config
        self.parallel_context = parallel_context
        self.transformer = BloomModel(config, parallel_context)
        tp = parallel_context.get_world_size(ParallelMode.TENSOR)
        # LM head: column-split over vocab, weights tied to the (vocab-sharded)
        # embedding; logits stay sharded — the parallel CE consumes them.
        self.lm_head = ColumnParallelLinear(
            config.hidden_size, config.vocab_size, bias=False,
            gather_output=False, sequence_parallel=config.sequence_parallel,
            parallel_context=parallel_context)
        self.lm_head.weight = self.transformer.word_embeddings.weight
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
            hidden, presents = self.transformer(input_ids, past=past,
                                                use_cache=True)
            return self.lm_head(hidden), presents
        hidden = self.transformer(input_ids)
        logits = self.lm_head(hidden)
        if labels is None:
            return logits
        shift_labels = _shifted_labels(labels)
        if self.loss_fn is not None:
            loss = self.loss_fn(logits, shift_labels)
        elif logits.is_cuda:
            from pipegoose_amd.ops.cross_entropy import fused_cross_entropy
            loss = fused_cross_entropy(
                logits.reshape(-1, logits.size(-1)), shift_labels.reshape(-1))
        else:
            loss = TF.cross_entropy(
                logits.float().reshape(-1, logits.size(-1)),
                shift_labels.reshape(-1), ignore_index=-100)
        return loss


    def new_kv_cache(self, batch_size: int, max_len: int):
        """Preallocated static KV cache for decode (models/kv_cache.py)."""
        from pipegoose_amd.models.kv_cache import StaticKVCache
        attn = self.transformer.h[0].self_attention
        p = next(self.parameters())
        return StaticKVCache(len(self.transformer.h), batch_size,
                             attn.num_heads, max_len, attn.head_dim,
                             p.dtype, p.device)

    def new_graph_kv_cache(self, batch_size: int, max_len: int):
        """Cache bank for hipGraph-captured decode (models/graph_decode.py)."""
        from pipegoose_amd.models.kv_cache import GraphKVCache
        attn = self.transformer.h[0].self_attention
        p = next(self.parameters())
        return GraphKVCache(len(self.transformer.h), batch_size,
                            attn.num_heads, max_len, attn.head_dim,
                            p.dtype, p.device)

    def gradient_checkpointing_enable(self, enabled: bool = True):
        """Recompute each block in backward instead of storing activations —
        trades ~30% step time for O(sqrt) activation memory (capability the
        reference lacked; composes with TP/SP/DP)."""
        self.transformer.gradient_checkpointing = enabled

    @torch.no_grad()
    def generate(self, input_ids, max_new_tokens: int = 20,
                 temperature: float = 0.0, top_k: int = 0,
                 eos_token_id=None):
        from pipegoose_amd.models.generation import generate as _generate
        return _generate(self, input_ids, max_new_tokens=max_new_tokens,
                         temperature=temperature, top_k=top_k,
                         parallel_context=self.parallel_context,
                         eos_token_id=eos_token_id)
