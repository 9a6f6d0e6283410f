"""Native MI355X model families (TP/SP-sharded by construction)."""
from pipegoose_amd.models.bloom import (BloomConfig, BloomForCausalLM,
                                        bloom_560m, bloom_1b7, bloom_7b1,
                                        bloom_tiny, make_causal_lm_loss)
from pipegoose_amd.models.llama import (LlamaConfig, LlamaForCausalLM,
                                        llama_1b, llama_tiny)
from pipegoose_amd.models.graph_decode import GraphDecoder
from pipegoose_amd.models.kv_cache import GraphKVCache, StaticKVCache

__all__ = [
    "BloomConfig", "BloomForCausalLM", "bloom_560m", "bloom_1b7",
    "bloom_7b1", "bloom_tiny", "make_causal_lm_loss",
    "LlamaConfig", "LlamaForCausalLM", "llama_1b", "llama_tiny",
    "GraphDecoder", "GraphKVCache", "StaticKVCache",
]
