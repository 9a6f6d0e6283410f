"""Gradient buckets: flat buffers for fused DP all-reduce.

Completes what the reference left unfinished (core/bucket/{bucket,dist}.py —
never flushed, ctor mismatch).  A bucket owns one contiguous buffer per
(dtype, group) pair; param grads are copied in and ``param.grad`` is re-pointed
at the bucket slice, so subsequent accumulation lands in the flat buffer and
the all-reduce needs no copy-back.

Sized by constants.BUCKET_SIZE_MB for the xGMI fabric (7 P2P links/GPU):
large enough to amortize RCCL launch, small enough to overlap with backward.
"""
from typing import Dict, List, Optional, Tuple

import torch
import torch.distributed as dist

from pipegoose_amd.constants import BUCKET_SIZE_MB
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class Bucket:
    def __init__(self, size: int, dtype: torch.dtype, device: torch.device):
        self.buffer = torch.zeros(size, dtype=dtype, device=device)
        self.offset = 0
        self.params: List[torch.nn.Parameter] = []
        self.is_closed = False

    @property
    def size(self) -> int:
        return self.buffer.numel()

    @property
    def available_size(self) -> int:
        return self.size - self.offset

    def is_full_with(self, numel: int) -> bool:
        return numel > self.available_size

    def add_grad(self, param: torch.nn.Parameter) -> torch.Tensor:
        """Copy param.grad into the bucket and re-point param.grad at the slice."""
        numel = param.grad.numel()
        assert numel <= self.available_size
        sl = self.buffer[self.offset:self.offset + numel]
        sl.copy_(param.grad.reshape(-1))
        param.grad = sl.view_as(param.grad)
        self.offset += numel
        self.params.append(param)
        return sl

    def filled_view(self) -> torch.Tensor:
        return self.buffer[: self.offset]

    def clear(self):
        self.offset = 0
        self.params.clear()
        self.is_closed = False


class BucketManager:
    """Per-(mode, dtype) buckets with async flush over the mode's group."""

    def __init__(self, parallel_context: ParallelContext,
                 bucket_size_mb: int = BUCKET_SIZE_MB):
        self.parallel_context = parallel_context
        self.bucket_bytes = bucket_size_mb * 1024 * 1024
        self.buckets: Dict[Tuple[ParallelMode, torch.dtype], Bucket] = {}
        self.pending_works: List[Tuple[object, torch.Tensor, int]] = []

    def _get_bucket(self, mode: ParallelMode, dtype: torch.dtype,
                    device: torch.device, min_numel: int) -> Bucket:
        key = (mode, dtype)
        numel = max(self.bucket_bytes // dtype.itemsize, min_numel)
        if key not in self.buckets:
            self.buckets[key] = Bucket(numel, dtype, device)
        return self.buckets[key]

    def add_param(self, param: torch.nn.Parameter, mode: ParallelMode):
        """Queue a param's grad; flush the bucket first if it would overflow."""
        numel = param.grad.numel()
        bucket = self._get_bucket(mode, param.grad.dtype, param.grad.device, numel)
        if bucket.is_full_with(numel):
            self.flush(mode, param.grad.dtype)
            bucket = self._get_bucket(mode, param.grad.dtype, param.grad.device, numel)
        bucket.add_grad(param)

    def flush(self, mode: Optional[ParallelMode] = None,
              dtype: Optional[torch.dtype] = None):
        """Launch async all-reduce on matching non-empty buckets."""
        for (m, dt), bucket in self.buckets.items():
            if mode is not None and m != mode:
                continue
            if dtype is not None and dt != dtype:
                continue
            if bucket.offset == 0:
                continue
            world = self.parallel_context.get_world_size(m)
            if world > 1:
                work = dist.all_reduce(
                    bucket.filled_view(),
                    group=self.parallel_context.get_group(m),
                    async_op=True,
                )
            else:
                work = None
            self.pending_works.append((work, bucket.filled_view(), world))
            # New bucket for further grads this step: re-point to fresh storage
            # so the in-flight buffer isn't overwritten.
            self.buckets[(m, dt)] = Bucket(bucket.size, dt, bucket.buffer.device)

    def wait_all(self):
        """Complete outstanding reduces and average (post-divide: better bf16
        precision than the reference's pre-divide, data_parallel.py:34-43)."""
        for work, view, world in self.pending_works:
            if work is not None:
                work.wait()
            if world > 1:
                view.div_(world)
        self.pending_works.clear()
