"""Typed P2P tensor transport.

Replaces the reference's 4-round-trip codec (pipegoose/distributed/_p2p.py:
dtype msg + requires_grad msg + shape msg + payload) with a single fixed-size
preamble + payload: 2 messages per tensor.  Over RCCL the preamble and payload
ride the same xGMI link back-to-back.
"""
from typing import Any

import torch
import torch.distributed as dist

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode

# dtype table (order is the wire format -- do not reorder)
ID_TO_DTYPE = [
    torch.bfloat16,
    torch.float16,
    torch.float32,
    torch.float64,
    torch.uint8,
    torch.int8,
    torch.int16,
    torch.int32,
    torch.int64,
    torch.bool,
]
DTYPE_TO_ID = {dtype: i for i, dtype in enumerate(ID_TO_DTYPE)}

_MAX_DIMS = 8
_HEADER_NUMEL = 3 + _MAX_DIMS  # [dtype_id, requires_grad, ndim, *shape]


class P2P:
    def __init__(self, parallel_context: ParallelContext,
                 parallel_mode: ParallelMode = ParallelMode.GLOBAL):
        self.parallel_context = parallel_context
        self.parallel_mode = parallel_mode

    def _comm_device(self) -> torch.device:
        # RCCL requires device tensors; gloo wants CPU.
        if dist.get_backend() == "nccl":
            return self.parallel_context.device
        return torch.device("cpu")

    def send(self, data: Any, dst: int, tag: int = 0):
        assert torch.is_tensor(data), f"P2P.send expects a tensor, got {type(data)}"
        device = self._comm_device()
        header = torch.zeros(_HEADER_NUMEL, dtype=torch.int64, device=device)
        header[0] = DTYPE_TO_ID[data.dtype]
        header[1] = int(data.requires_grad)
        header[2] = data.dim()
        for i, s in enumerate(data.shape):
            header[3 + i] = s
        dist.send(header, dst=dst, tag=tag)
        payload = data.detach().contiguous().to(device)
        dist.send(payload, dst=dst, tag=tag)

    def send_many(self, tensors, dst: int, tag: int = 0):
        """Typed send of a tuple/list of tensors: one count preamble, then
        the per-tensor typed codec (used by the pipeline engine's first
        microbatch when a stage boundary carries multiple values)."""
        device = self._comm_device()
        n = torch.tensor([len(tensors)], dtype=torch.int64, device=device)
        dist.send(n, dst=dst, tag=tag)
        for t in tensors:
            self.send(t, dst, tag=tag)

    def recv_many(self, src: int, tag: int = 0):
        device = self._comm_device()
        n = torch.zeros(1, dtype=torch.int64, device=device)
        dist.recv(n, src=src, tag=tag)
        return [self.recv(src, tag=tag) for _ in range(int(n.item()))]

    def recv(self, src: int, tag: int = 0) -> torch.Tensor:
        device = self._comm_device()
        header = torch.zeros(_HEADER_NUMEL, dtype=torch.int64, device=device)
        dist.recv(header, src=src, tag=tag)
        header_cpu = header.tolist()
        dtype = ID_TO_DTYPE[header_cpu[0]]
        requires_grad = bool(header_cpu[1])
        ndim = header_cpu[2]
        shape = header_cpu[3:3 + ndim]
        payload = torch.empty(shape, dtype=dtype, device=device)
        dist.recv(payload, src=src, tag=tag)
        payload.requires_grad_(requires_grad and payload.is_floating_point())
        return payload
