"""ZeRO-1 DistributedOptimizer: sharded step == non-sharded reference step
(reference: tests/optim/zero/test_optim.py:15-50)."""
import torch
from torch import nn

from pipegoose_amd.nn import DataParallel
from pipegoose_amd.optim import DistributedOptimizer
from pipegoose_amd.optim.sharding import OptimizerStateSharding
from pipegoose_amd.testing import init_parallel_context, spawn


def _model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(8, 32), nn.GELU(), nn.Linear(32, 4))


def test_sharding_is_balanced_and_complete():
    model = _model()
    ctx = type("FakeCtx", (), {})()

    class Fake:
        def get_world_size(self, mode):
            return 2

    sharder = OptimizerStateSharding(
        [{"params": list(model.parameters()), "lr": 0.1}], Fake(), None)
    parts = sharder.shard()
    all_params = [p for rank in parts for g in rank for p in g["params"]]
    assert len(all_params) == len(list(model.parameters()))
    numels = [sum(p.numel() for g in rank for p in g["params"]) for rank in parts]
    assert all(n > 0 for n in numels)


def run_zero_step(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=world_size)
    model = DataParallel(_model(), ctx).parallelize()
    optim = DistributedOptimizer(
        torch.optim.Adam(model.parameters(), lr=1e-2), ctx)

    # each local optimizer holds fewer params than the full model
    n_local = sum(len(g["params"]) for g in optim.optim.param_groups)
    n_total = len(list(model.parameters()))
    assert n_local < n_total

    # reference: full Adam on averaged grads
    ref_model = _model()
    ref_optim = torch.optim.Adam(ref_model.parameters(), lr=1e-2)

    for step in range(3):
        xs = []
        for r in range(world_size):
            torch.manual_seed(500 + step * world_size + r)
            xs.append(torch.randn(4, 8))

        optim.zero_grad()
        model(xs[rank]).pow(2).mean().backward()
        optim.step()

        ref_optim.zero_grad()
        losses = [ref_model(x).pow(2).mean() for x in xs]
        (sum(losses) / world_size).backward()
        ref_optim.step()

    for p, p_ref in zip(model.parameters(), ref_model.parameters()):
        assert torch.allclose(p, p_ref, atol=1e-5), (p - p_ref).abs().max()
    ctx.destroy()


def test_zero1_matches_full_adam():
    spawn(run_zero_step, world_size=2)
