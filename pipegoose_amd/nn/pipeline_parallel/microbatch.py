"""Microbatch splitting (reference microbatch.py had torch.split size/count
confusion — SURVEY known-bug list; torch.chunk here)."""
from typing import Dict, List, Union

import torch


def split(inputs: Union[torch.Tensor, Dict[str, torch.Tensor]],
          n_microbatches: int) -> List:
    if torch.is_tensor(inputs):
        assert inputs.size(0) % n_microbatches == 0, (
            f"batch {inputs.size(0)} not divisible by {n_microbatches} microbatches"
        )
        return list(torch.chunk(inputs, n_microbatches, dim=0))
    assert isinstance(inputs, dict)
    keys = list(inputs.keys())
    chunked = {k: torch.chunk(v, n_microbatches, dim=0) for k, v in inputs.items()}
    return [{k: chunked[k][i] for k in keys} for i in range(n_microbatches)]
