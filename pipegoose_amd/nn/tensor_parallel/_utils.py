"""Vocab-shard index math (reference: nn/tensor_parallel/_utils.py:4-14)."""


class VocabUtility:
    @staticmethod
    def get_vocab_range_from_partition_size(partition_size: int, rank: int):
        start = rank * partition_size
        return start, start + partition_size

    @staticmethod
    def get_vocab_range_from_global_vocab_size(world_size: int, rank: int, vocab_size: int):
        assert vocab_size % world_size == 0
        partition = vocab_size // world_size
        return VocabUtility.get_vocab_range_from_partition_size(partition, rank)
