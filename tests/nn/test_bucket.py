"""Gradient-bucket substrate (the reference's core/bucket was unfinished:
never flushed, ctor mismatch — SURVEY.md §2.3).  Covers the semantics its
tests/core/bucket/test_bucket.py checked, plus the flush/async behavior the
reference never reached.
"""
import torch
from torch import nn

from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.data_parallel.bucket import Bucket, BucketManager
from pipegoose_amd.testing.utils import init_parallel_context, spawn
from pipegoose_amd.utils.memory import get_tensor_storage_mem_loc


def _param_with_grad(shape):
    p = nn.Parameter(torch.randn(shape))
    p.grad = torch.randn(shape)
    return p


def test_bucket_add_grad_repoints_storage():
    torch.manual_seed(0)
    b = Bucket(100, torch.float32, torch.device("cpu"))
    p = _param_with_grad((4, 5))
    orig = p.grad.clone()
    b.add_grad(p)
    # grad now lives INSIDE the bucket buffer (no copy-back needed later)
    assert get_tensor_storage_mem_loc(p.grad) == \
        get_tensor_storage_mem_loc(b.buffer)
    assert torch.equal(p.grad, orig)
    assert b.offset == 20 and b.available_size == 80


def test_bucket_capacity_accounting():
    b = Bucket(10, torch.float32, torch.device("cpu"))
    assert not b.is_full_with(10)
    assert b.is_full_with(11)
    p = _param_with_grad((6,))
    b.add_grad(p)
    assert b.is_full_with(5)
    b.clear()
    assert b.offset == 0 and b.available_size == 10


def _run_manager_averages(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=2)
    mgr = BucketManager(ctx, bucket_size_mb=1)
    torch.manual_seed(10 + rank)
    params = [_param_with_grad((8, 8)) for _ in range(3)]
    expected = []
    for p in params:
        g = p.grad.clone()
        import torch.distributed as dist
        dist.all_reduce(g)
        expected.append(g / world_size)
    for p in params:
        mgr.add_param(p, ParallelMode.DATA)
    mgr.flush()
    mgr.wait_all()
    for p, e in zip(params, expected):
        assert torch.allclose(p.grad, e, atol=1e-6)
    ctx.destroy()


def test_bucket_manager_all_reduce_average_dp2():
    spawn(_run_manager_averages, world_size=2)


def _run_overflow_flush(rank, world_size, port):
    """Grads bigger than the bucket trigger an intermediate flush; all data
    still arrives averaged."""
    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=2)
    mgr = BucketManager(ctx, bucket_size_mb=1)  # 262144 fp32 elements
    torch.manual_seed(20 + rank)
    big = [_param_with_grad((200_000,)) for _ in range(3)]  # forces flushes
    import torch.distributed as dist
    expected = []
    for p in big:
        g = p.grad.clone()
        dist.all_reduce(g)
        expected.append(g / world_size)
    for p in big:
        mgr.add_param(p, ParallelMode.DATA)
    mgr.flush()
    mgr.wait_all()
    for p, e in zip(big, expected):
        assert torch.allclose(p.grad, e, atol=1e-6)
    ctx.destroy()


def test_bucket_manager_overflow_flush_dp2():
    spawn(_run_overflow_flush, world_size=2)


def _run_param_bigger_than_bucket(rank, world_size, port):
    """A single grad LARGER than the bucket itself (bloom-560m's embedding
    grad is ~500 MB vs the 25 MB bucket) must get a dedicated buffer —
    previously the post-flush bucket kept the old size and asserted."""
    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=2)
    mgr = BucketManager(ctx, bucket_size_mb=1)  # 262144 fp32 elements
    torch.manual_seed(30 + rank)
    small = _param_with_grad((64,))
    huge = _param_with_grad((400_000,))  # > bucket capacity
    import torch.distributed as dist
    expected = []
    for p in (small, huge):
        g = p.grad.clone()
        dist.all_reduce(g)
        expected.append(g / world_size)
    mgr.add_param(small, ParallelMode.DATA)
    mgr.add_param(huge, ParallelMode.DATA)
    mgr.flush()
    mgr.wait_all()
    for p, e in zip((small, huge), expected):
        assert torch.allclose(p.grad, e, atol=1e-6)
    ctx.destroy()


def test_bucket_param_bigger_than_bucket_dp2():
    spawn(_run_param_bigger_than_bucket, world_size=2)
