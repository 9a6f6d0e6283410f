"""Typed P2P codec: 2-message preamble+payload protocol (reference's
distributed/_p2p.py used 4 round trips)."""
import pytest
import torch

from pipegoose_amd.distributed.p2p import DTYPE_TO_ID, ID_TO_DTYPE, P2P
from pipegoose_amd.testing.utils import init_parallel_context, spawn


def test_dtype_table_roundtrip():
    assert len(ID_TO_DTYPE) == 10
    for i, dt in enumerate(ID_TO_DTYPE):
        assert DTYPE_TO_ID[dt] == i


def _run_p2p(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    p2p = P2P(ctx)
    cases = [
        torch.randn(3, 4),
        torch.randn(2, 2, 2, dtype=torch.float64),
        torch.arange(7, dtype=torch.int64),
        torch.tensor([True, False, True]),
        torch.randn(5, requires_grad=True),
        torch.randn(1, 2048, 64).to(torch.bfloat16),
    ]
    torch.manual_seed(42)  # both ranks build identical reference tensors
    if rank == 0:
        for t in cases:
            p2p.send(t, dst=1)
    else:
        for t in cases:
            got = p2p.recv(src=0)
            assert got.dtype == t.dtype
            assert got.shape == t.shape
            assert torch.equal(got, t.detach())
            assert got.requires_grad == (t.requires_grad and t.is_floating_point())
    ctx.destroy()


def test_p2p_typed_send_recv():
    spawn(_run_p2p, world_size=2)
