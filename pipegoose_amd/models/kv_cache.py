"""Static KV cache for incremental decode.

The naive cache re-`cat`s the whole [B, H, S, D] history every step and
layer (O(S) copies per token).  A StaticKVCache preallocates
[B, H, max_len, D] once and index-writes the new step — no growth copies,
stable addresses (hipGraph-friendly for a future captured decode step).

Beyond-reference capability (the reference had no decode path at all).
"""
from typing import List, Tuple

import torch


class LayerKVCache:
    def __init__(self, B: int, H: int, max_len: int, D: int,
                 dtype: torch.dtype, device: torch.device):
        self.k = torch.empty(B, H, max_len, D, dtype=dtype, device=device)
        self.v = torch.empty(B, H, max_len, D, dtype=dtype, device=device)
        self.len = 0

    def append(self, k_new: torch.Tensor, v_new: torch.Tensor
               ) -> Tuple[torch.Tensor, torch.Tensor]:
        """Write the new steps and return views of the filled prefix."""
        s = k_new.size(2)
        assert self.len + s <= self.k.size(2), "KV cache overflow"
        self.k[:, :, self.len:self.len + s] = k_new
        self.v[:, :, self.len:self.len + s] = v_new
        self.len += s
        return self.k[:, :, :self.len], self.v[:, :, :self.len]


class StaticKVCache:
    """Per-layer static caches; duck-types the (k, v) tuple list the model's
    ``past=`` plumbing expects via __getitem__."""

    def __init__(self, n_layers: int, B: int, H: int, max_len: int, D: int,
                 dtype: torch.dtype, device):
        self.layers: List[LayerKVCache] = [
            LayerKVCache(B, H, max_len, D, dtype, torch.device(device))
            for _ in range(n_layers)]

    def __getitem__(self, i: int) -> "LayerKVCache":
        return self.layers[i]

    def __len__(self):
        return len(self.layers)

    @property
    def seq_len(self) -> int:
        return self.layers[0].len
