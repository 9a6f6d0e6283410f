"""Collective primitives over spawned gloo processes (reference:
tests/distributed/test_functional.py:64-219)."""
import torch

from pipegoose_amd.distributed import ParallelMode
from pipegoose_amd.distributed import functional as F
from pipegoose_amd.testing import init_parallel_context, spawn

MODE = ParallelMode.GLOBAL


def run_all_reduce(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    x = torch.full((4,), float(rank + 1))
    F.all_reduce(x, parallel_context=ctx, parallel_mode=MODE)
    expected = sum(r + 1 for r in range(world_size))
    assert torch.equal(x, torch.full((4,), float(expected)))
    ctx.destroy()


def run_all_gather(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    x = torch.full((2, 3), float(rank))
    out = F.all_gather(x, dim=0, parallel_context=ctx, parallel_mode=MODE)
    assert out.shape == (2 * world_size, 3)
    for r in range(world_size):
        assert torch.equal(out[2 * r:2 * r + 2], torch.full((2, 3), float(r)))
    # dim=-1 path
    out = F.all_gather(x, dim=-1, parallel_context=ctx, parallel_mode=MODE)
    assert out.shape == (2, 3 * world_size)
    ctx.destroy()


def run_reduce_scatter(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    x = torch.arange(4 * world_size, dtype=torch.float32)
    out = F.reduce_scatter(x, dim=0, parallel_context=ctx, parallel_mode=MODE)
    expected = torch.arange(4 * rank, 4 * rank + 4, dtype=torch.float32) * world_size
    assert torch.equal(out, expected)
    ctx.destroy()


def run_all_to_all(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    # rank r sends chunk i (filled with r*ws+i) to rank i
    x = torch.cat([torch.full((2,), float(rank * world_size + i)) for i in range(world_size)])
    out = F.all_to_all(x, in_dim=0, out_dim=0, parallel_context=ctx, parallel_mode=MODE)
    expected = torch.cat([torch.full((2,), float(r * world_size + rank)) for r in range(world_size)])
    assert torch.equal(out, expected)
    ctx.destroy()


def run_broadcast_scatter(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    x = torch.full((3,), float(rank))
    F.broadcast(x, src=0, parallel_context=ctx, parallel_mode=MODE)
    assert torch.equal(x, torch.zeros(3))
    y = torch.arange(2 * world_size, dtype=torch.float32)
    shard = F.scatter(y, dim=0, parallel_context=ctx, parallel_mode=MODE)
    assert torch.equal(shard, torch.arange(2 * rank, 2 * rank + 2, dtype=torch.float32))
    F.barrier(ctx, MODE)
    ctx.destroy()


def run_send_recv(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    data = torch.arange(6, dtype=torch.bfloat16).reshape(2, 3)
    F.send(data, src=0, dst=1, parallel_context=ctx)
    received = F.recv(src=0, dst=1, parallel_context=ctx)
    if rank == 1:
        assert received.dtype == torch.bfloat16
        assert torch.equal(received.float(), data.float())
    else:
        assert received is None
    ctx.destroy()


def test_all_reduce():
    spawn(run_all_reduce, world_size=2)


def test_all_gather():
    spawn(run_all_gather, world_size=2)


def test_reduce_scatter():
    spawn(run_reduce_scatter, world_size=2)


def test_all_to_all():
    spawn(run_all_to_all, world_size=2)


def test_broadcast_scatter_barrier():
    spawn(run_broadcast_scatter, world_size=2)


def test_send_recv():
    spawn(run_send_recv, world_size=2)


def run_variable_a2a(rank, world_size, port):
    from pipegoose_amd.distributed import functional as F
    ctx = init_parallel_context(rank, world_size, port)
    # rank 0 sends [1, 2] rows; rank 1 sends [3, 0] rows
    in_splits = [1, 2] if rank == 0 else [3, 0]
    torch.manual_seed(rank)
    x = torch.arange(sum(in_splits) * 4, dtype=torch.float32).reshape(-1, 4) \
        + 100 * rank
    out, out_splits = F.all_to_all_variable(x, in_splits, ctx, MODE)
    if rank == 0:
        assert out_splits == [1, 3]
        assert torch.equal(out[:1], x[:1])                       # own slice
        assert torch.equal(out[1:], torch.arange(12, dtype=torch.float32)
                           .reshape(3, 4) + 100)                 # rank1's 3
    else:
        assert out_splits == [2, 0]
        assert torch.equal(out, torch.arange(4, 12, dtype=torch.float32)
                           .reshape(2, 4))                       # rank0's 2
    # exchange_splits alone
    assert F.exchange_splits(in_splits, ctx, MODE) == out_splits
    ctx.destroy()


def test_all_to_all_variable_uneven_with_zero():
    spawn(run_variable_a2a, world_size=2)
