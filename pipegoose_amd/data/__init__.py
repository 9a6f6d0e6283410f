"""Data utilities for 4D-parallel training.

The reference shipped no data layer (its tests pulled HF datasets ad hoc);
training at scale needs the sharding rules to be explicit:

- samples are sharded over the DATA group ONLY — every rank in the same
  TP×PP block must see the identical batch or the parallel math is wrong;
- the synthetic LM dataset mirrors what bench.py trains on (random token
  ids of the benchmark shape) so examples/tests run with no network.
"""
from typing import Optional

import torch
from torch.utils.data import DataLoader, Dataset, DistributedSampler

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class SyntheticLMDataset(Dataset):
    """Deterministic random token sequences: item i is seeded by (seed, i),
    so every rank materializes identical data without communication."""

    def __init__(self, n_samples: int, seq_len: int, vocab_size: int,
                 seed: int = 0):
        self.n_samples = n_samples
        self.seq_len = seq_len
        self.vocab_size = vocab_size
        self.seed = seed

    def __len__(self):
        return self.n_samples

    def __getitem__(self, idx):
        g = torch.Generator().manual_seed(self.seed * 1_000_003 + idx)
        ids = torch.randint(0, self.vocab_size, (self.seq_len,), generator=g)
        return {"input_ids": ids, "labels": ids.clone()}


def build_dataloader(
    dataset: Dataset,
    micro_batch_size: int,
    parallel_context: Optional[ParallelContext] = None,
    shuffle: bool = True,
    seed: int = 0,
    drop_last: bool = True,
    num_workers: int = 0,
) -> DataLoader:
    """DataLoader sharded over the DATA group: DP replicas get disjoint
    sample shards; all TP/PP ranks of one replica get the same shard."""
    ctx = parallel_context or ParallelContext.get_context()
    dp = ctx.get_world_size(ParallelMode.DATA) if ctx is not None else 1
    dp_rank = ctx.get_local_rank(ParallelMode.DATA) if ctx is not None else 0
    sampler = DistributedSampler(
        dataset, num_replicas=dp, rank=dp_rank, shuffle=shuffle, seed=seed,
        drop_last=drop_last) if dp > 1 else None
    return DataLoader(
        dataset, batch_size=micro_batch_size, sampler=sampler,
        shuffle=(shuffle and sampler is None), drop_last=drop_last,
        num_workers=num_workers)
