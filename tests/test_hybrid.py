"""Hybrid 3D parallel integration: TP x PP x DP on gloo
(reference: tests/test_hybrid.py oracle pattern)."""
import torch
import torch.nn.functional as TF

from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
from pipegoose_amd.nn import DataParallel
from pipegoose_amd.nn.pipeline_parallel import PipelineParallel
from pipegoose_amd.optim import DistributedOptimizer
from pipegoose_amd.testing import init_parallel_context, spawn


def lm_loss(logits, labels):
    shift_logits = logits[:, :-1].reshape(-1, logits.size(-1)).float()
    shift_labels = labels[:, 1:].reshape(-1)
    return TF.cross_entropy(shift_logits, shift_labels)


def run_tp_pp(rank, world_size, port):
    """TP2 x PP2 (world 4): loss parity with the single-process model."""
    ctx = init_parallel_context(rank, world_size, port,
                               tensor_parallel_size=2, pipeline_parallel_size=2)
    torch.manual_seed(321)
    model = BloomForCausalLM(bloom_tiny(), ctx)  # TP-aware by construction

    torch.manual_seed(8)
    ids = torch.randint(0, 256, (4, 16))

    # tp=2 -> logits are vocab-sharded; use the vocab-parallel CE
    from pipegoose_amd.nn.tensor_parallel import VocabParallelCrossEntropy
    vp_ce = VocabParallelCrossEntropy(parallel_context=ctx)

    def sharded_lm_loss(logits, labels):
        return vp_ce(logits[:, :-1].contiguous(), labels[:, 1:].contiguous())

    pp = PipelineParallel(model, ctx, n_microbatches=2,
                          loss_fn=sharded_lm_loss).parallelize()
    loss = pp(ids, ids)

    from pipegoose_amd.distributed.parallel_mode import ParallelMode
    if ctx.is_last_rank(ParallelMode.PIPELINE):
        assert loss is not None and torch.isfinite(loss)
    ctx.destroy()


def test_tp2_pp2_runs():
    spawn(run_tp_pp, world_size=4)


def run_dp_pp(rank, world_size, port):
    """PP2 x DP2 (world 4): replicas stay identical after a synced step."""
    ctx = init_parallel_context(rank, world_size, port, pipeline_parallel_size=2)
    torch.manual_seed(11)
    model = BloomForCausalLM(bloom_tiny(), ctx)
    pp = PipelineParallel(model, ctx, n_microbatches=2, loss_fn=lm_loss).parallelize()
    pp = DataParallel(pp, ctx).parallelize()
    optim = DistributedOptimizer(torch.optim.Adam(pp.parameters(), lr=1e-3), ctx)

    from pipegoose_amd.distributed.parallel_mode import ParallelMode
    dp_rank = ctx.get_local_rank(ParallelMode.DATA)
    for step in range(2):
        torch.manual_seed(900 + step * 10 + dp_rank)
        ids = torch.randint(0, 256, (4, 16))
        optim.zero_grad()
        pp(ids, ids)  # engine runs fwd+bwd; DP syncs at the tail
        optim.step()

    # all DP replicas of this pipeline stage must hold identical params
    flat = torch.cat([p.detach().reshape(-1) for p in pp.parameters()])
    peers = [torch.empty_like(flat) for _ in range(2)]
    torch.distributed.all_gather(peers, flat, group=ctx.get_group(ParallelMode.DATA))
    assert torch.allclose(peers[0], peers[1], atol=1e-6)
    ctx.destroy()


def test_pp2_dp2_replicas_stay_synced():
    spawn(run_dp_pp, world_size=4)


def run_tp_pp_sp(rank, world_size, port):
    """Config 5 shape: TP2 x PP2 with sequence parallelism — loss parity
    with the non-SP hybrid (same seeds, same data)."""
    import torch
    from pipegoose_amd.distributed.parallel_mode import ParallelMode
    from pipegoose_amd.models.bloom import (BloomConfig, BloomForCausalLM,
                                            make_causal_lm_loss)
    from pipegoose_amd.nn.pipeline_parallel import PipelineParallel
    from pipegoose_amd.testing.utils import init_parallel_context

    ctx = init_parallel_context(rank, world_size, port,
                                tensor_parallel_size=2,
                                pipeline_parallel_size=2)

    def build(sp):
        cfg = BloomConfig(vocab_size=256, hidden_size=64, n_layer=2, n_head=4,
                          sequence_parallel=sp)
        torch.manual_seed(12)
        model = BloomForCausalLM(cfg, ctx)
        return PipelineParallel(model, ctx, n_microbatches=2,
                                loss_fn=make_causal_lm_loss(ctx)).parallelize()

    torch.manual_seed(13)
    ids = torch.randint(0, 256, (4, 16))

    loss_ref = build(sp=False)(ids, ids)
    loss_sp = build(sp=True)(ids, ids)
    if ctx.is_last_rank(ParallelMode.PIPELINE):
        assert loss_ref is not None and loss_sp is not None
        assert torch.allclose(loss_ref, loss_sp, atol=1e-5), \
            (loss_ref, loss_sp)
    ctx.destroy()


def test_tp2_pp2_sequence_parallel_matches():
    spawn(run_tp_pp_sp, world_size=4)


def run_dp_interleaved(rank, world_size, port):
    """Interleaved PP2(v2) x DP2: deferred DP sync at the engine tail keeps
    replicas identical after a step (mirror of run_dp_pp)."""
    import dataclasses
    from pipegoose_amd.distributed.parallel_mode import ParallelMode

    ctx = init_parallel_context(rank, world_size, port, pipeline_parallel_size=2)
    cfg = dataclasses.replace(bloom_tiny(), n_layer=4)
    torch.manual_seed(12)
    model = BloomForCausalLM(cfg, ctx)
    pp = PipelineParallel(model, ctx, n_microbatches=2,
                          schedule="interleaved", virtual_stages=2,
                          loss_fn=lm_loss).parallelize()
    pp = DataParallel(pp, ctx).parallelize()
    optim = DistributedOptimizer(torch.optim.Adam(pp.parameters(), lr=1e-3), ctx)

    dp_rank = ctx.get_local_rank(ParallelMode.DATA)
    for step in range(2):
        torch.manual_seed(910 + step * 10 + dp_rank)
        ids = torch.randint(0, 256, (4, 16))
        optim.zero_grad()
        pp(ids, ids)
        optim.step()

    flat = torch.cat([p.detach().reshape(-1) for p in pp.parameters()])
    peers = [torch.empty_like(flat) for _ in range(2)]
    torch.distributed.all_gather(peers, flat, group=ctx.get_group(ParallelMode.DATA))
    assert torch.allclose(peers[0], peers[1], atol=1e-6)
    ctx.destroy()


def test_interleaved_pp2_dp2_replicas_stay_synced():
    spawn(run_dp_interleaved, world_size=4)
