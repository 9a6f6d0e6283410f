"""ProfileByMemory feeding non-uniform pipeline cuts (reference
partitioning/profile.py:19-49, there unused by the partitioner)."""
import torch
from torch import nn

from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
from pipegoose_amd.partitioning import ProfileByMemory
from pipegoose_amd.nn.pipeline_parallel.partitioner import UniformPartitioner
from pipegoose_amd.testing.utils import init_parallel_context, spawn


def _run(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port)
    torch.manual_seed(0)
    model = BloomForCausalLM(bloom_tiny(), ctx)
    sizes = ProfileByMemory(model).profile(torch.randint(0, 256, (1, 8)))
    assert len(sizes) == bloom_tiny().n_layer
    assert all(s > 0 for s in sizes)

    # sizes drive the cut: skew block 0 to be huge → stage 0 gets 1 block
    stages = UniformPartitioner(model, ctx, sizes=[100.0, 1.0]).split(n_partitions=2)
    assert len(stages) == 2
    ctx.destroy()


def test_profile_by_memory_cpu():
    spawn(_run, world_size=1)


def test_profile_plain_sequential():
    model = nn.Sequential(nn.Linear(8, 32), nn.ReLU(), nn.Linear(32, 8))
    sizes = ProfileByMemory(model, device=torch.device("cpu")).profile(
        torch.randn(4, 8))
    assert len(sizes) == 3
