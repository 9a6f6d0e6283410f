"""Context-parallel axis: CONTEXT groups + ring attention in the model families.

cp2 over gloo: each rank holds an S/cp sequence shard of the SAME tiny
native model; logits must match the full-sequence single-model oracle's
shard, and after backward + CONTEXT-group grad sync the param grads must
match the oracle's (parameters replicate over CP like DP).
"""
import pytest
import torch

from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn import DataParallel
from pipegoose_amd.testing.utils import init_parallel_context, spawn


def test_context_group_layout():
    from pipegoose_amd.distributed._initializers import (
        ContextParallelGroupInitializer)

    class _NoGroup(ContextParallelGroupInitializer):
        pass

    # layout math only (no process groups): tp2 cp2 dp2 pp1, world 8
    import pipegoose_amd.distributed._initializers as I
    captured = []
    orig = I._make_group

    def fake(rank, ranks, mode):
        captured.append(list(ranks))
        return None

    I._make_group = fake
    try:
        ContextParallelGroupInitializer(0, 8, 2, 1, 2, 2).init_dist_group()
    finally:
        I._make_group = orig
    assert captured == [[0, 2], [1, 3], [4, 6], [5, 7]]


def _bloom_tiny(cp, ctx):
    from pipegoose_amd.models.bloom import BloomConfig, BloomForCausalLM
    torch.manual_seed(31)
    cfg = BloomConfig(vocab_size=128, hidden_size=32, n_layer=2, n_head=4,
                      context_parallel=cp)
    return BloomForCausalLM(cfg, ctx)


def _llama_tiny(cp, ctx):
    from pipegoose_amd.models.llama import LlamaConfig, LlamaForCausalLM
    torch.manual_seed(33)
    cfg = LlamaConfig(vocab_size=128, hidden_size=32, intermediate_size=64,
                      n_layer=2, n_head=4, n_kv_head=2, context_parallel=cp)
    return LlamaForCausalLM(cfg, ctx)


def _run_model_cp(rank, world_size, port, family):
    ctx = init_parallel_context(rank, world_size, port,
                                context_parallel_size=world_size)
    build = _bloom_tiny if family == "bloom" else _llama_tiny
    model = build(True, ctx)
    DataParallel(model, ctx, mode=ParallelMode.CONTEXT).parallelize()

    ref = build(False, ctx)  # same seed -> same weights
    torch.manual_seed(7)
    B, S = 2, 8 * world_size
    ids = torch.randint(0, 128, (B, S))
    ref_logits = ref(ids).float()
    # per-token CE over the FULL sequence, mean
    ref_loss = torch.nn.functional.cross_entropy(
        ref_logits.reshape(-1, 128), ids.reshape(-1))
    ref_loss.backward()

    Sl = S // world_size
    sl = slice(rank * Sl, (rank + 1) * Sl)
    logits = model(ids[:, sl]).float()
    assert torch.allclose(logits, ref_logits[:, sl], atol=1e-4), \
        (logits - ref_logits[:, sl]).abs().max()

    # shard-mean CE; the global mean is the average of shard means here
    # (equal shard sizes), so scale by 1/cp and let the CONTEXT all-reduce
    # (which averages) restore the global-mean gradient
    loss = torch.nn.functional.cross_entropy(
        logits.reshape(-1, 128), ids[:, sl].reshape(-1))
    loss.backward()

    ref_named = dict(ref.named_parameters())
    for name, p in model.named_parameters():
        if p.grad is None:
            continue
        want = ref_named[name].grad
        assert want is not None, name
        assert torch.allclose(p.grad, want, atol=1e-4), \
            (name, (p.grad - want).abs().max())
    ctx.destroy()


@pytest.mark.parametrize("family", ["bloom", "llama"])
def test_model_cp2_logits_and_grad_parity(family):
    spawn(_run_model_cp, world_size=2, family=family)
