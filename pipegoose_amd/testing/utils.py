"""Test harness: simulate multi-rank runs with OS processes on localhost.

Same trick as the reference (pipegoose/testing/utils.py:20-63): spawn N
processes with torch.multiprocessing, rendezvous on 127.0.0.1 with the gloo
backend — real process groups and collectives, CPU-only.  GPU tests use the
same harness with backend="nccl" (RCCL) and are marked @pytest.mark.gpu.
"""
import os
import socket
from functools import partial
from typing import Callable

import torch
import torch.multiprocessing as mp

from pipegoose_amd.distributed.parallel_context import ParallelContext


def find_free_port() -> int:
    with socket.socket(socket.AF_INET, socket.SOCK_STREAM) as s:
        s.bind(("127.0.0.1", 0))
        return s.getsockname()[1]


def _wrapper(rank: int, func: Callable, world_size: int, port: int, kwargs: dict):
    # Each spawned process must look like a torchrun rank.
    os.environ["RANK"] = str(rank)
    os.environ["LOCAL_RANK"] = str(rank)
    os.environ["WORLD_SIZE"] = str(world_size)
    os.environ["MASTER_ADDR"] = "127.0.0.1"
    os.environ["MASTER_PORT"] = str(port)
    func(rank=rank, world_size=world_size, port=port, **kwargs)


def spawn(func: Callable, world_size: int = 1, timeout: float = 180.0, **kwargs):
    """Run ``func(rank, world_size, port, **kwargs)`` in ``world_size`` processes."""
    port = find_free_port()
    mp.start_processes(
        partial(_wrapper, func=func, world_size=world_size, port=port, kwargs=kwargs),
        nprocs=world_size,
        start_method="spawn",
        join=True,
    )


def init_parallel_context(
    rank: int,
    world_size: int,
    port: int,
    tensor_parallel_size: int = 1,
    pipeline_parallel_size: int = 1,
    data_parallel_size: int = None,
    backend: str = "gloo",
    seed: int = 69,
    context_parallel_size: int = 1,
) -> ParallelContext:
    return ParallelContext(
        rank=rank,
        local_rank=rank,
        world_size=world_size,
        local_world_size=world_size,
        host="127.0.0.1",
        port=port,
        backend=backend,
        seed=seed,
        tensor_parallel_size=tensor_parallel_size,
        pipeline_parallel_size=pipeline_parallel_size,
        data_parallel_size=data_parallel_size,
        context_parallel_size=context_parallel_size,
    )


def calculate_parameter_similarity(module1: torch.nn.Module, module2: torch.nn.Module,
                                   rtol: float = 1e-3) -> float:
    """Fraction of parameters that are allclose between two modules."""
    n_same, n_total = 0, 0
    for p1, p2 in zip(module1.parameters(), module2.parameters()):
        n_total += 1
        if p1.shape == p2.shape and torch.allclose(p1, p2, rtol=rtol):
            n_same += 1
    return n_same / max(n_total, 1)
