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
    """Per-(mode, dtype, owner) buckets with async flush over the mode's group.

    ``owner`` is None for plain data parallelism (all-reduce everywhere) or a
    ZeRO shard-owner local rank (params tagged ``_zero_owner`` by
    DistributedOptimizer(grad_reduce="shard")): those buckets are REDUCED to
    the owner only — half the wire bytes of an all-reduce, and exactly what
    ZeRO-1 needs since only the owner steps those params."""

    def __init__(self, parallel_context: ParallelContext,
                 bucket_size_mb: int = None):
        if bucket_size_mb is None:
            # RuntimeConfig (env PG_BUCKET_MB / yaml) overrides the constant
            from pipegoose_amd.config import get_config
            bucket_size_mb = get_config().bucket_size_mb or BUCKET_SIZE_MB
        self.parallel_context = parallel_context
        self.bucket_bytes = bucket_size_mb * 1024 * 1024
        self.buckets: Dict[Tuple[ParallelMode, torch.dtype, Optional[int]],
                           Bucket] = {}
        self.pending_works: List[Tuple[object, torch.Tensor, int, bool]] = []

    def _get_bucket(self, key, device: torch.device, min_numel: int) -> Bucket:
        numel = max(self.bucket_bytes // key[1].itemsize, min_numel)
        if key not in self.buckets:
            self.buckets[key] = Bucket(numel, key[1], device)
        return self.buckets[key]

    def add_param(self, param: torch.nn.Parameter, mode: ParallelMode):
        """Queue a param's grad; flush the bucket first if it would overflow."""
        numel = param.grad.numel()
        owner = getattr(param, "_zero_owner", None)             if mode == ParallelMode.DATA else None
        key = (mode, param.grad.dtype, owner)
        bucket = self._get_bucket(key, param.grad.device, numel)
        if bucket.is_full_with(numel):
            self._flush_key(key)
            bucket = self._get_bucket(key, param.grad.device, numel)
            if bucket.size < numel:
                # a param bigger than the bucket (e.g. a vocab embedding's
                # grad is hundreds of MB) gets a dedicated full-size buffer
                bucket = self.buckets[key] = Bucket(numel, key[1],
                                                    param.grad.device)
        bucket.add_grad(param)

    def _flush_key(self, key):
        mode, dt, owner = key
        bucket = self.buckets.get(key)
        if bucket is None or bucket.offset == 0:
            return
        world = self.parallel_context.get_world_size(mode)
        my_rank = self.parallel_context.get_local_rank(mode)
        work = None
        if world > 1:
            group = self.parallel_context.get_group(mode)
            if owner is None:
                work = dist.all_reduce(bucket.filled_view(), group=group,
                                       async_op=True)
            else:
                dst = self.parallel_context.get_ranks_in_group(mode)[owner]
                work = dist.reduce(bucket.filled_view(), dst=dst, group=group,
                                   async_op=True)
        needs_divide = world > 1 and (owner is None or owner == my_rank)
        self.pending_works.append(
            (work, bucket.filled_view(), world, needs_divide))
        # New bucket for further grads this step: re-point to fresh storage
        # so the in-flight buffer isn't overwritten.
        self.buckets[key] = Bucket(bucket.size, dt, bucket.buffer.device)

    def flush(self, mode: Optional[ParallelMode] = None,
              dtype: Optional[torch.dtype] = None):
        """Launch async reduces on matching non-empty buckets."""
        for key in list(self.buckets.keys()):
            if mode is not None and key[0] != mode:
                continue
            if dtype is not None and key[1] != dtype:
                continue
            self._flush_key(key)

    def wait_all(self):
        """Complete outstanding reduces and average (post-divide: better bf16
        precision than the reference's pre-divide, data_parallel.py:34-43).
        Owner-routed buckets only divide on the owner (elsewhere the buffer
        holds partial sums nobody reads)."""
        for work, view, world, needs_divide in self.pending_works:
            if work is not None:
                work.wait()
            if needs_divide:
                view.div_(world)
        self.pending_works.clear()
