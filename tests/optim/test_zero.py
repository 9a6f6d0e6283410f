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


def _run_shard_reduce_parity(rank, world_size, port):
    """grad_reduce="shard" training == grad_reduce="replicate" training."""
    import torch
    from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
    from pipegoose_amd.nn import DataParallel
    from pipegoose_amd.optim import DistributedOptimizer
    from pipegoose_amd.testing.utils import init_parallel_context
    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=2)

    def train(mode):
        torch.manual_seed(7)
        model = BloomForCausalLM(bloom_tiny(), ctx)
        model = DataParallel(model, ctx).parallelize()
        optim = DistributedOptimizer(
            torch.optim.Adam(model.parameters(), lr=1e-3), ctx,
            grad_reduce=mode)
        for step in range(3):
            torch.manual_seed(100 + 10 * step + rank)
            ids = torch.randint(0, 256, (2, 12))
            optim.zero_grad()
            model(ids, labels=ids).backward()
            optim.clip_grad_norm_(1.0)
            optim.step()
        return [p.detach().clone() for p in model.parameters()]

    ref = train("replicate")
    got = train("shard")
    for a, b in zip(ref, got):
        assert torch.allclose(a, b, atol=1e-5), (a - b).abs().max()
    ctx.destroy()


def test_zero_shard_grad_reduce_matches_replicate_dp2():
    from pipegoose_amd.testing.utils import spawn
    spawn(_run_shard_reduce_parity, world_size=2)


def _run_flat_shard_rebuild(rank, world_size, port):
    """Params are views into persistent flat shard buffers; a model move
    (.to()) re-allocates storages and must trigger a transparent rebuild."""
    import torch
    from torch import nn
    from pipegoose_amd.optim import DistributedOptimizer
    from pipegoose_amd.testing.utils import init_parallel_context

    ctx = init_parallel_context(rank, world_size, port, data_parallel_size=2)
    torch.manual_seed(4)
    model = nn.Sequential(nn.Linear(16, 16), nn.Linear(16, 16))
    optim = DistributedOptimizer(torch.optim.SGD(model.parameters(), lr=0.1),
                                 ctx)
    assert optim._flat_valid()
    # simulate a post-setup move: re-allocate every param storage
    for p in model.parameters():
        p.data = p.data.clone()
    assert not optim._flat_valid()

    x = torch.randn(4, 16)
    model(x).sum().backward()
    for p in model.parameters():  # emulate DP grad sync (identical data)
        torch.distributed.all_reduce(p.grad)
    optim.step()
    assert optim._flat_valid()  # rebuilt
    # replicas identical after the in-place broadcast
    for p in model.parameters():
        peers = [torch.empty_like(p.data) for _ in range(world_size)]
        torch.distributed.all_gather(peers, p.data)
        assert torch.equal(peers[0], peers[1])
    ctx.destroy()


def test_zero_flat_shard_views_and_rebuild():
    from pipegoose_amd.testing.utils import spawn
    spawn(_run_flat_shard_rebuild, world_size=2)
