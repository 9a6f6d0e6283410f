"""TP role mapping (reference tests/nn/tensor_parallel/test_parallel_mapping.py)."""
from pipegoose_amd.nn.tensor_parallel.parallel_mapping import (
    Column, Row, TensorParallelMapping)
from pipegoose_amd.nn.tensor_parallel._utils import VocabUtility


def test_bloom_mapping_roles():
    M = TensorParallelMapping
    assert M.is_column_parallel("transformer.h.0.self_attention.query_key_value")
    assert M.is_column_parallel("transformer.h.3.mlp.dense_h_to_4h")
    assert M.is_row_parallel("transformer.h.0.self_attention.dense")
    assert M.is_row_parallel("transformer.h.11.mlp.dense_4h_to_h")
    assert M.is_lm_head("lm_head")
    assert not M.is_column_parallel("transformer.word_embeddings")
    assert not M.is_row_parallel("transformer.h.0.input_layernorm")


def test_llama_mapping_roles():
    M = TensorParallelMapping
    assert M.is_column_parallel("model.layers.5.self_attn.q_proj")
    assert M.is_column_parallel("model.layers.0.mlp.gate_proj")
    assert M.is_row_parallel("model.layers.2.self_attn.o_proj")
    assert M.is_row_parallel("model.layers.9.mlp.down_proj")


def test_register_new_architecture():
    M = TensorParallelMapping
    M.register("my-arch", [Column("attn.qkv"), Row("attn.out")])
    assert M.is_column_parallel("blocks.0.attn.qkv")
    assert M.is_row_parallel("blocks.7.attn.out")


def test_vocab_utility_ranges():
    assert VocabUtility.get_vocab_range_from_partition_size(100, 0) == (0, 100)
    assert VocabUtility.get_vocab_range_from_partition_size(100, 3) == (300, 400)
    assert VocabUtility.get_vocab_range_from_global_vocab_size(4, 1, 400) == (100, 200)
