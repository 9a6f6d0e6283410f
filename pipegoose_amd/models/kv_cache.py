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


class GraphLayerKVCache(LayerKVCache):
    """LayerKVCache whose decode write is shape-static: in graph mode
    ``append`` index-writes at a DEVICE position tensor and returns the FULL
    buffers, so a captured hipGraph replays with constant shapes/addresses
    while the position advances as data (models/graph_decode.py)."""

    def __init__(self, B, H, max_len, D, dtype, device, pos_t: torch.Tensor):
        super().__init__(B, H, max_len, D, dtype, device)
        # full-length attention reads UNFILLED slots (masked by -inf bias);
        # they must be finite — inf garbage in q·k would make inf + (-inf)
        # = NaN and poison the whole softmax row
        self.k.zero_()
        self.v.zero_()
        self.pos_t = pos_t          # int64 [1], shared across layers
        self.graph_mode = False

    def append(self, k_new: torch.Tensor, v_new: torch.Tensor):
        if not self.graph_mode:     # eager prefill: normal growing prefix
            return super().append(k_new, v_new)
        # decode step: k_new is [B, H, 1, D]; write at pos_t, return full S_max
        self.k.index_copy_(2, self.pos_t, k_new)
        self.v.index_copy_(2, self.pos_t, v_new)
        return self.k, self.v


class GraphKVCache:
    """Static cache bank for graph-captured decode: one shared device
    position scalar, per-layer static buffers.  ``graph_mode`` off = behaves
    like StaticKVCache (eager prefill fills the prefix); on = every layer
    writes at ``pos_t`` and attention runs full-length with a position mask."""

    def __init__(self, n_layers: int, B: int, H: int, max_len: int, D: int,
                 dtype: torch.dtype, device):
        device = torch.device(device)
        self.pos_t = torch.zeros(1, dtype=torch.long, device=device)
        self.max_len = max_len
        self.layers: List[GraphLayerKVCache] = [
            GraphLayerKVCache(B, H, max_len, D, dtype, device, self.pos_t)
            for _ in range(n_layers)]

    def __getitem__(self, i: int) -> GraphLayerKVCache:
        return self.layers[i]

    def __len__(self):
        return len(self.layers)

    @property
    def seq_len(self) -> int:
        return self.layers[0].len

    def enter_graph_mode(self):
        """Call after prefill: freeze shapes, position = filled length."""
        self.pos_t.fill_(self.layers[0].len)
        for layer in self.layers:
            layer.graph_mode = True

    def advance(self):
        self.pos_t.add_(1)  # capture-safe: device-side increment

    def reset(self):
        """Back to eager-prefill state (buffer addresses unchanged, so a
        previously captured graph stays valid)."""
        self.pos_t.zero_()
        for layer in self.layers:
            layer.graph_mode = False
            layer.len = 0
