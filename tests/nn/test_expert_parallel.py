"""Expert parallelism: routers, context/loss bookkeeping, mask dispatch
(reference parity, nn/expert_parallel/experts.py:41-102), and the all-to-all
dispatch path (MI355X EP=N config) checked against the mask path as oracle.
"""
import pytest
import torch
from torch import nn

from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
from pipegoose_amd.nn.expert_parallel import (ExpertContext, ExpertLayer,
                                              ExpertLoss, ExpertParallel,
                                              SwitchNoisePolicy, Top1Router,
                                              Top2Router)
from pipegoose_amd.nn.expert_parallel.experts import Experts
from pipegoose_amd.testing.utils import init_parallel_context, spawn

H, NUM_EXPERTS = 16, 4


# --------------------------------------------------------------- pure logic

def test_top1_router_shapes_and_losses():
    torch.manual_seed(0)
    router = Top1Router(SwitchNoisePolicy(), NUM_EXPERTS, H)
    router.eval()
    x = torch.randn(2, 6, H)
    out = router(x)
    assert out.dispatch_order.shape == (12,)
    assert out.dispatch_order.max() < NUM_EXPERTS
    assert out.weight.shape == (12, NUM_EXPERTS)
    assert out.aux_loss.item() >= 1.0 - 1e-5  # lower bound at perfect balance
    assert out.z_loss.item() >= 0


def test_top2_router_two_experts_per_token():
    torch.manual_seed(0)
    router = Top2Router(SwitchNoisePolicy(), NUM_EXPERTS, H)
    router.eval()
    out = router(torch.randn(3, 5, H))
    assert out.dispatch_order.shape == (15, 2)
    # the two chosen experts differ for every token
    assert (out.dispatch_order[:, 0] != out.dispatch_order[:, 1]).all()


def test_router_capacity_truncates():
    torch.manual_seed(0)
    router = Top1Router(SwitchNoisePolicy(), 2, H, expert_capacity=(0.5, 0.5))
    router.eval()
    out = router(torch.randn(1, 8, H))
    # capacity 0.5 * 8 / 2 = 2 tokens per expert at most survive the mask
    assert (out.weight > 0).sum() <= 4


def test_expert_context_push_pop():
    ctx = ExpertContext.get_instance()
    ctx.pop_all_aux_loss(), ctx.pop_all_z_loss()  # drain
    ctx.push_aux_loss(torch.tensor(1.0))
    ctx.push_z_loss(torch.tensor(2.0))
    assert [t.item() for t in ctx.pop_all_aux_loss()] == [1.0]
    assert ctx.pop_all_aux_loss() == []
    assert [t.item() for t in ctx.pop_all_z_loss()] == [2.0]


def test_expert_loss_adds_scaled_router_losses():
    ctx = ExpertContext.get_instance()
    ctx.pop_all_aux_loss(), ctx.pop_all_z_loss()
    loss_fn = ExpertLoss(lambda a, b: (a - b).pow(2).mean(), aux_weight=0.01,
                         z_weight=0.1)
    ctx.push_aux_loss(torch.tensor(3.0))
    ctx.push_z_loss(torch.tensor(5.0))
    base = (torch.ones(2) - torch.zeros(2)).pow(2).mean()
    total = loss_fn(torch.ones(2), torch.zeros(2))
    assert torch.allclose(total, base + 0.01 * 3.0 + 0.1 * 5.0)
    assert ctx.aux_losses == [] and ctx.z_losses == []


# ------------------------------------------------------- single-rank expert

def _mlp():
    return nn.Sequential(nn.Linear(H, 4 * H), nn.GELU(), nn.Linear(4 * H, H))


def _run_single_rank_expert_layer(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(1)
    layer = ExpertLayer(NUM_EXPERTS, _mlp(),
                        Top1Router(SwitchNoisePolicy(), NUM_EXPERTS, H),
                        enable_tensor_parallel=False, parallel_context=ctx)
    layer.eval()
    x = torch.randn(2, 6, H)
    out = layer(x)
    assert out.shape == x.shape
    # every token went through exactly one expert: recompute manually
    ExpertContext.get_instance().pop_all_aux_loss()
    ExpertContext.get_instance().pop_all_z_loss()
    ctx.destroy()


def test_expert_layer_single_rank():
    spawn(_run_single_rank_expert_layer, world_size=1)


# ------------------------------------------- tp2: mask vs all-to-all parity

def _run_dispatch_parity(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(2)  # same seed every rank -> identical experts
    proto = _mlp()

    mask_experts = Experts(NUM_EXPERTS, proto, enable_tensor_parallel=True,
                           parallel_context=ctx, dispatch="mask")
    torch.manual_seed(2)
    a2a_experts = Experts(NUM_EXPERTS, proto, enable_tensor_parallel=True,
                          parallel_context=ctx, dispatch="alltoall")
    for p1, p2 in zip(mask_experts.parameters(), a2a_experts.parameters()):
        assert torch.equal(p1, p2)

    torch.manual_seed(3)
    x = torch.randn(2, 8, H, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    route = torch.randint(0, NUM_EXPERTS, (2 * 8,))

    out_mask = mask_experts(x, route)
    out_a2a = a2a_experts(x2, route)
    assert torch.allclose(out_mask, out_a2a, atol=1e-6), \
        (out_mask - out_a2a).abs().max()

    g = torch.randn_like(out_mask)
    out_mask.backward(g)
    out_a2a.backward(g)
    # mask path leaves per-rank PARTIAL input grads (summed later by TP);
    # the a2a path yields the full replicated grad on every rank.
    import torch.distributed as dist
    mask_grad_sum = x.grad.clone()
    dist.all_reduce(mask_grad_sum)
    assert torch.allclose(mask_grad_sum, x2.grad, atol=1e-5), \
        (mask_grad_sum - x2.grad).abs().max()
    for p1, p2 in zip(mask_experts.parameters(), a2a_experts.parameters()):
        if p1.grad is None:
            assert p2.grad is None or p2.grad.abs().max() == 0
        else:
            assert torch.allclose(p1.grad, p2.grad, atol=1e-6)
    ctx.destroy()


def test_alltoall_dispatch_matches_mask_tp2():
    spawn(_run_dispatch_parity, world_size=2)


def _run_uneven_routing(rank, world_size, port):
    """All tokens to one expert: extreme split imbalance must still work."""
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(4)
    experts = Experts(NUM_EXPERTS, _mlp(), enable_tensor_parallel=True,
                      parallel_context=ctx, dispatch="alltoall")
    x = torch.randn(1, 6, H)
    route = torch.full((6,), 3, dtype=torch.long)  # expert 3 lives on rank 1
    out = experts(x, route)
    ref = experts.experts[1](x.reshape(-1, H)) if rank == 1 else None
    # oracle: rank 1's local expert idx 1 == global expert 3
    torch.manual_seed(4)
    oracle = Experts(NUM_EXPERTS, _mlp(), enable_tensor_parallel=True,
                     parallel_context=ctx, dispatch="mask")
    assert torch.allclose(out, oracle(x, route), atol=1e-6)
    ctx.destroy()


def test_alltoall_uneven_routing_tp2():
    spawn(_run_uneven_routing, world_size=2)


# ----------------------------------------------------------- model surgery

def _run_surgery(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(5)
    model = BloomForCausalLM(bloom_tiny(), ctx)
    cfg = bloom_tiny()
    ep = ExpertParallel(
        model, NUM_EXPERTS,
        router=Top1Router(SwitchNoisePolicy(), NUM_EXPERTS, cfg.hidden_size),
        parallel_context=ctx)
    model = ep.parallelize()
    n_expert_layers = sum(isinstance(m, ExpertLayer) for m in model.modules())
    assert n_expert_layers == cfg.n_layer
    ids = torch.randint(0, 256, (2, 10))
    loss_fn = ExpertLoss(lambda logits, labels: logits.float().mean())
    logits = model(ids)
    loss = loss_fn(logits, ids)
    loss.backward()
    router_grads = [p.grad for m in model.modules() if isinstance(m, ExpertLayer)
                    for p in m.router.parameters()]
    assert any(g is not None and g.abs().sum() > 0 for g in router_grads), \
        "router got no gradient (aux/z losses should reach the gate)"
    ctx.destroy()


def test_expert_parallel_surgery_and_backward():
    spawn(_run_surgery, world_size=1)


# ------------------------------------------------- hybrid EP x TP x DP (w4)

def _run_hybrid_ep(rank, world_size, port):
    """tp2 x dp2: experts sharded over TP, expert grads reduced over
    EXPERT_DATA, dense grads over DATA (reference
    tests/nn/expert_parallel/test_hybrid_expert_parallel.py:58-80)."""
    import torch.distributed as dist
    from pipegoose_amd.distributed.parallel_mode import ParallelMode
    from pipegoose_amd.nn import DataParallel

    ctx = init_parallel_context(rank, world_size, port,
                                tensor_parallel_size=2, data_parallel_size=2)
    cfg = bloom_tiny()
    torch.manual_seed(6)  # identical init everywhere
    model = BloomForCausalLM(cfg, ctx)
    dense_expert = nn.Sequential(
        nn.Linear(cfg.hidden_size, 4 * cfg.hidden_size), nn.GELU(),
        nn.Linear(4 * cfg.hidden_size, cfg.hidden_size))
    model = ExpertParallel(
        model, NUM_EXPERTS, expert=dense_expert,
        router=Top1Router(SwitchNoisePolicy(), NUM_EXPERTS, cfg.hidden_size),
        enable_tensor_parallel=True, dispatch="alltoall",
        parallel_context=ctx).parallelize()
    model = DataParallel(model, ctx).parallelize()

    torch.manual_seed(100 + ctx.get_local_rank(ParallelMode.DATA))
    ids = torch.randint(0, 256, (2, 8))
    loss_fn = ExpertLoss(lambda logits, labels: logits.float().pow(2).mean())
    model.zero_grad()
    # routers must agree across ranks: eval disables jitter noise
    model.eval()
    loss = loss_fn(model(ids), ids)
    loss.backward()

    expert_params = [p for p in model.parameters()
                     if getattr(p, "is_expert", False) and p.grad is not None]
    assert expert_params, "no expert grads"
    for p in expert_params[:3]:
        g = p.grad.flatten()[:16].clone()
        peers = [torch.empty_like(g) for _ in
                 range(ctx.get_world_size(ParallelMode.EXPERT_DATA))]
        dist.all_gather(peers, g, group=ctx.get_group(ParallelMode.EXPERT_DATA))
        for peer in peers:
            assert torch.allclose(g, peer, atol=1e-6), "expert grad not synced"
    ctx.destroy()


def test_hybrid_ep_tp_dp_world4():
    spawn(_run_hybrid_ep, world_size=4)


def test_noise_policy_identical_across_identical_states():
    """Same-seeded policies draw identical noise (the TP-sync guarantee)."""
    torch.manual_seed(1)  # global seed must NOT influence the policy
    n1 = SwitchNoisePolicy(seed=7)
    torch.manual_seed(2)
    n2 = SwitchNoisePolicy(seed=7)
    x = torch.zeros(4, 8)
    a = n1.sample_like(x)
    b = n2.sample_like(x)
    assert torch.equal(a, b)
    assert (a >= 0.9).all() and (a <= 1.1).all()


# --------------------------- top-2 + capacity: weighted a2a vs mask parity

def _run_weighted_dispatch_parity(rank, world_size, port, k):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(4)
    proto = _mlp()

    mask_experts = Experts(NUM_EXPERTS, proto, enable_tensor_parallel=True,
                           parallel_context=ctx, dispatch="mask")
    torch.manual_seed(4)
    a2a_experts = Experts(NUM_EXPERTS, proto, enable_tensor_parallel=True,
                          parallel_context=ctx, dispatch="alltoall")

    torch.manual_seed(5)
    N = 2 * 8
    x = torch.randn(2, 8, H, requires_grad=True)
    x2 = x.detach().clone().requires_grad_(True)
    if k == 1:
        route = torch.randint(0, NUM_EXPERTS, (N,))
        route2d = route.unsqueeze(-1)
    else:
        route = torch.stack([torch.randperm(NUM_EXPERTS)[:k]
                             for _ in range(N)])  # distinct experts/token
        route2d = route
    # routing weights with some capacity-dropped entries (zeros)
    weight = torch.zeros(N, NUM_EXPERTS)
    wvals = torch.rand(N, k) + 0.1
    weight.scatter_(1, route2d, wvals)
    dropped = torch.rand(N) < 0.25
    weight[dropped] = 0.0  # fully dropped tokens

    out_mask = mask_experts(x, route, weight)
    out_a2a = a2a_experts(x2, route, weight)
    assert torch.allclose(out_mask, out_a2a, atol=1e-5), \
        (out_mask - out_a2a).abs().max()
    # dropped tokens produce zeros (residual-only pass-through)
    flat = out_a2a.reshape(N, H)
    assert flat[dropped].abs().max() == 0

    g = torch.randn_like(out_mask)
    out_mask.backward(g)
    out_a2a.backward(g)
    import torch.distributed as dist
    mask_grad_sum = x.grad.clone()
    dist.all_reduce(mask_grad_sum)
    assert torch.allclose(mask_grad_sum, x2.grad, atol=1e-5)
    for p1, p2 in zip(mask_experts.parameters(), a2a_experts.parameters()):
        if p1.grad is not None and p2.grad is not None:
            assert torch.allclose(p1.grad, p2.grad, atol=1e-5)
    ctx.destroy()


def test_weighted_top1_capacity_a2a_matches_mask():
    spawn(_run_weighted_dispatch_parity, world_size=2, k=1)


def test_weighted_top2_a2a_matches_mask():
    """Top-2 combine weights on the wire path: each
    token ships to both experts, gate-weighted sum, capacity drops never
    sent."""
    spawn(_run_weighted_dispatch_parity, world_size=2, k=2)


def _run_top2_single_rank_oracle(rank, world_size, port):
    """Weighted top-2 combine vs a hand-computed oracle (no parallelism)."""
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(6)
    experts = Experts(NUM_EXPERTS, _mlp(), enable_tensor_parallel=False,
                      parallel_context=ctx)
    N = 10
    x = torch.randn(1, N, H)
    route = torch.stack([torch.randperm(NUM_EXPERTS)[:2] for _ in range(N)])
    weight = torch.zeros(N, NUM_EXPERTS)
    weight.scatter_(1, route, torch.rand(N, 2) + 0.1)
    out = experts(x, route, weight).reshape(N, H)
    for t in range(N):
        ref = sum(weight[t, e] * experts.experts[e](x[0, t:t + 1]).squeeze(0)
                  for e in route[t].tolist())
        assert torch.allclose(out[t], ref, atol=1e-5)
    ctx.destroy()


def test_top2_weighted_combine_oracle():
    spawn(_run_top2_single_rank_oracle, world_size=1)


def _run_local_grouped_parity(rank, world_size, port):
    """PG_MOE_GROUPED=1 routes the mask path through the batched expert
    bank; outputs/grads must match the per-expert loop exactly."""
    import os
    os.environ["PG_MOE_GROUPED"] = "1"
    try:
        ctx = init_parallel_context(rank, world_size, port)
        torch.manual_seed(8)
        grouped = Experts(NUM_EXPERTS, _mlp(), enable_tensor_parallel=False,
                          parallel_context=ctx)
        torch.manual_seed(8)
        loop = Experts(NUM_EXPERTS, _mlp(), enable_tensor_parallel=False,
                       parallel_context=ctx)
        loop._grouped = None  # force the per-expert loop
        N = 24
        x1 = torch.randn(2, N // 2, H, requires_grad=True)
        x2 = x1.detach().clone().requires_grad_(True)
        route = torch.stack([torch.randperm(NUM_EXPERTS)[:2]
                             for _ in range(N)])
        weight = torch.zeros(N, NUM_EXPERTS)
        weight.scatter_(1, route, torch.rand(N, 2) + 0.1)
        o1 = grouped(x1, route, weight)
        o2 = loop(x2, route, weight)
        assert torch.allclose(o1, o2, atol=1e-5), (o1 - o2).abs().max()
        g = torch.randn_like(o1)
        o1.backward(g)
        o2.backward(g)
        assert torch.allclose(x1.grad, x2.grad, atol=1e-5)
        for p1, p2 in zip(grouped.parameters(), loop.parameters()):
            if p2.grad is None:
                continue
            assert torch.allclose(p1.grad, p2.grad, atol=1e-5)
        ctx.destroy()
    finally:
        os.environ.pop("PG_MOE_GROUPED", None)


def test_local_grouped_matches_loop():
    spawn(_run_local_grouped_parity, world_size=1)
