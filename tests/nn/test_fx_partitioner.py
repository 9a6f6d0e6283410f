"""FxUniformPartitioner: arbitrary inter-block dataflow through the pipeline.

The structural resolvers reject models without a pre/blocks/post shape; the
fx fallback traces, splits by parameter bytes, and threads every
cross-shard value (reference parity: its fx partitioner + cross-shard value
propagation, pipegoose/nn/pipeline_parallel/partitioner.py:129-219 — but
without the transformer-block-boundary restriction).  The skip connection
here makes the stage boundary a TUPLE, exercising the engine's multi-value
transport end to end with grad parity vs the unpartitioned model.
"""
import pytest
import torch
from torch import nn

from pipegoose_amd.nn.pipeline_parallel.engine import PipelineEngine
from pipegoose_amd.nn.pipeline_parallel.partitioner import (
    FxUniformPartitioner,
    UniformPartitioner,
)
from pipegoose_amd.testing.utils import init_parallel_context, spawn


class SkipNet(nn.Module):
    """Early activation consumed again late: not pre/blocks/post-shaped."""

    def __init__(self):
        super().__init__()
        torch.manual_seed(77)
        self.l1 = nn.Linear(16, 16)
        self.l2 = nn.Linear(16, 16)
        self.l3 = nn.Linear(16, 16)
        self.l4 = nn.Linear(16, 4)

    def forward(self, x):
        a = torch.relu(self.l1(x))
        b = torch.relu(self.l2(a))
        c = torch.relu(self.l3(b)) + a
        return self.l4(c)


def test_structural_resolver_rejects_skipnet():
    class FakeCtx:
        def get_world_size(self, m):
            return 2

    with pytest.raises(ValueError):
        UniformPartitioner(SkipNet(), FakeCtx()).split(2)


@pytest.mark.parametrize("pp", [2, 3])
def test_fx_split_local_parity(pp):
    class FakeCtx:
        def get_world_size(self, m):
            return pp

        def get_local_rank(self, m):
            return 0

    model = SkipNet()
    stages = FxUniformPartitioner(model, FakeCtx()).split(pp)
    x = torch.randn(4, 16)
    ref = model(x)
    h = x
    for st in stages:
        h = st(*h) if isinstance(h, tuple) else st(h)
    assert torch.allclose(h, ref, atol=1e-6)


def _run_fx_pipeline(rank, world_size, port, tp, pp, dp):
    ctx = init_parallel_context(rank, world_size, port, tp, pp, dp)
    torch.manual_seed(5)
    model = SkipNet()
    stage = FxUniformPartitioner(model, ctx).get_model_partition()
    loss_fn = lambda out, tgt: torch.nn.functional.mse_loss(out, tgt)  # noqa
    engine = PipelineEngine(stage, ctx, n_microbatches=4, loss_fn=loss_fn)

    torch.manual_seed(9)
    x = torch.randn(8, 16)
    y = torch.randn(8, 4)
    loss = engine.run(x, y)

    # oracle on every rank: same model, plain backward
    ref_model = SkipNet()
    ref_out = ref_model(x)
    chunks = ref_out.chunk(4)
    ychunks = y.chunk(4)
    ref_loss = sum(
        torch.nn.functional.mse_loss(c, yc) / 4
        for c, yc in zip(chunks, ychunks))
    ref_loss.backward()

    if rank == world_size - 1:
        assert torch.allclose(loss, ref_loss, atol=1e-5), (loss, ref_loss)

    # grad parity for the params this stage owns
    ref_named = dict(ref_model.named_modules())
    for name, p in stage.named_parameters():
        # submod params keep their original names under the split module
        leaf = name.split(".")[-1]            # weight|bias
        owner = name.split(".")[-2]           # l1..l4
        ref_p = getattr(ref_named[owner], leaf)
        assert ref_p.grad is not None
        assert torch.allclose(p.grad, ref_p.grad, atol=1e-5), \
            (name, (p.grad - ref_p.grad).abs().max())
    ctx.destroy()


@pytest.mark.parametrize("pp", [2, 3])
def test_fx_pipeline_engine_grad_parity(pp):
    spawn(_run_fx_pipeline, world_size=pp, tp=1, pp=pp, dp=1)
