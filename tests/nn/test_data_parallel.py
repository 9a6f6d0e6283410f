"""DataParallel: bucketed grad all-reduce equals averaged reference grads."""
import torch
from torch import nn

from pipegoose_amd.nn import DataParallel
from pipegoose_amd.testing import init_parallel_context, spawn


def _model(seed=0):
    torch.manual_seed(seed)
    return nn.Sequential(nn.Linear(8, 16), nn.GELU(), nn.Linear(16, 4))


def run_dp_grads(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=world_size)
    model = _model()
    model = DataParallel(model, ctx).parallelize()

    torch.manual_seed(100 + rank)  # different data per replica
    x = torch.randn(6, 8)
    loss = model(x).sum()
    loss.backward()

    # reference: average of per-rank grads computed in one process
    ref = _model()
    grads = []
    for r in range(world_size):
        torch.manual_seed(100 + r)
        xr = torch.randn(6, 8)
        ref.zero_grad()
        ref(xr).sum().backward()
        grads.append([p.grad.clone() for p in ref.parameters()])
    avg = [torch.stack(gs).mean(0) for gs in zip(*grads)]

    for p, g_ref in zip(model.parameters(), avg):
        assert torch.allclose(p.grad, g_ref, atol=1e-6), (p.grad - g_ref).abs().max()
    ctx.destroy()


def test_dp_grad_averaging():
    spawn(run_dp_grads, world_size=2)


def run_dp_multi_step(rank, world_size, port):
    """Buckets must stay correct across several fwd/bwd/step cycles."""
    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=world_size)
    model = DataParallel(_model(), ctx).parallelize()
    optim = torch.optim.SGD(model.parameters(), lr=0.05)
    for step in range(3):
        torch.manual_seed(1000 + step * world_size + rank)
        x = torch.randn(4, 8)
        optim.zero_grad()
        model(x).pow(2).mean().backward()
        optim.step()
    # replicas must remain identical after synced updates
    flat = torch.cat([p.detach().reshape(-1) for p in model.parameters()])
    gathered = [torch.empty_like(flat) for _ in range(world_size)]
    torch.distributed.all_gather(gathered, flat)
    for g in gathered:
        assert torch.allclose(flat, g, atol=1e-6)
    ctx.destroy()


def test_dp_multi_step():
    spawn(run_dp_multi_step, world_size=2)
