"""TP role mapping for HF architectures (reference:
nn/tensor_parallel/parallel_mapping.py + nn/parallel_mapping.py).

Ships bloom + albert + gpt2-style entries; extensible via
``TensorParallelMapping.register``.
"""
from pipegoose_amd.nn.parallel_mapping import ParallelInfo, ParallelMapping


class Column(ParallelInfo):
    pass


class Row(ParallelInfo):
    pass


class LMHead(ParallelInfo):
    pass


class FusedColumn(ParallelInfo):
    """Column-parallel layer whose output dim stacks n projections BLOCKWISE
    (GPT-2 ``c_attn``: [Q|K|V] along the out dim) — each block must be sliced
    separately, a plain chunk would hand rank 0 all of Q."""

    def __init__(self, *names, n: int = 3, **kwargs):
        super().__init__(*names, **kwargs)
        self.n = n


class TensorParallelMapping(ParallelMapping):
    __MAPPING__ = {
        "bloom-560m": [
            Column("mlp.dense_h_to_4h", "self_attention.query_key_value"),
            Row("mlp.dense_4h_to_h", "self_attention.dense"),
            LMHead("lm_head"),
        ],
        "albert-base-v2": [
            Column("attention.query", "attention.key", "attention.value", "ffn"),
            Row("attention.dense", "ffn_output"),
        ],
        "llama": [
            Column("self_attn.q_proj", "self_attn.k_proj", "self_attn.v_proj",
                   "mlp.gate_proj", "mlp.up_proj"),
            Row("self_attn.o_proj", "mlp.down_proj"),
            LMHead("lm_head"),
        ],
        "gpt2": [
            FusedColumn("attn.c_attn", n=3),  # Conv1D, blockwise q|k|v
            Column("mlp.c_fc"),
            Row("attn.c_proj", "mlp.c_proj"),
            LMHead("lm_head"),
        ],
        # pipegoose_amd native models (models/{bloom,llama}.py) use the same
        # names as the HF architectures.
    }

    @classmethod
    def is_column_parallel(cls, module_name: str) -> bool:
        return isinstance(cls._search(module_name), Column)

    @classmethod
    def is_row_parallel(cls, module_name: str) -> bool:
        return isinstance(cls._search(module_name), Row)

    @classmethod
    def is_lm_head(cls, module_name: str) -> bool:
        return isinstance(cls._search(module_name), LMHead)

    @classmethod
    def is_fused_column(cls, module_name: str) -> bool:
        return isinstance(cls._search(module_name), FusedColumn)

    @classmethod
    def get_fused_count(cls, module_name: str) -> int:
        info = cls._search(module_name)
        return info.n if isinstance(info, FusedColumn) else 1
