"""TP layers vs single-process reference (oracle pattern, reference:
tests/nn/tensor_parallel/test_linear.py etc.)."""
import torch
import torch.nn.functional as TF
from torch import nn

from pipegoose_amd.nn.tensor_parallel.embedding import ParallelEmbedding
from pipegoose_amd.nn.tensor_parallel.linear import ColumnParallelLinear, RowParallelLinear
from pipegoose_amd.nn.tensor_parallel.loss import VocabParallelCrossEntropy
from pipegoose_amd.testing import init_parallel_context, spawn

IN, OUT, BATCH = 16, 24, 5


def _ref_linear(seed=42):
    torch.manual_seed(seed)
    ref = nn.Linear(IN, OUT)
    x = torch.randn(BATCH, IN, requires_grad=True)
    return ref, x


def run_column_linear(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=world_size)
    ref, x = _ref_linear()
    y_ref = ref(x)

    layer = ColumnParallelLinear(IN, OUT, gather_output=True, parallel_context=ctx)
    layer.weight.data = ref.weight.data.chunk(world_size, dim=0)[rank].clone()
    layer.bias.data = ref.bias.data.chunk(world_size, dim=0)[rank].clone()

    x2 = x.detach().clone().requires_grad_(True)
    y = layer(x2)
    assert torch.allclose(y, y_ref, atol=1e-5), (y - y_ref).abs().max()

    # backward: input grads must match the non-parallel reference
    y_ref.sum().backward()
    y.sum().backward()
    assert torch.allclose(x2.grad, x.grad, atol=1e-5)
    # weight grad is this rank's slice of the reference weight grad
    wg_ref = ref.weight.grad.chunk(world_size, dim=0)[rank]
    assert torch.allclose(layer.weight.grad, wg_ref, atol=1e-5)
    ctx.destroy()


def run_row_linear(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=world_size)
    ref, x = _ref_linear(7)
    y_ref = ref(x)

    layer = RowParallelLinear(IN, OUT, parallel_context=ctx)
    layer.weight.data = ref.weight.data.chunk(world_size, dim=1)[rank].clone()
    layer.bias.data = ref.bias.data.clone()

    x2 = x.detach().clone().requires_grad_(True)
    y = layer(x2)
    assert torch.allclose(y, y_ref, atol=1e-5), (y - y_ref).abs().max()

    y_ref.sum().backward()
    y.sum().backward()
    assert torch.allclose(x2.grad, x.grad, atol=1e-5)
    wg_ref = ref.weight.grad.chunk(world_size, dim=1)[rank]
    assert torch.allclose(layer.weight.grad, wg_ref, atol=1e-5)
    ctx.destroy()


def run_embedding(rank, world_size, port):
    VOCAB, DIM = 32, 8
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=world_size)
    torch.manual_seed(3)
    ref = nn.Embedding(VOCAB, DIM)
    ids = torch.randint(0, VOCAB, (4, 6))
    y_ref = ref(ids)

    layer = ParallelEmbedding(VOCAB, DIM, parallel_context=ctx)
    layer.weight.data = ref.weight.data.chunk(world_size, dim=0)[rank].clone()
    y = layer(ids)
    assert torch.allclose(y, y_ref, atol=1e-5)

    y_ref.sum().backward()
    y.sum().backward()
    wg_ref = ref.weight.grad.chunk(world_size, dim=0)[rank]
    assert torch.allclose(layer.weight.grad, wg_ref, atol=1e-5)
    ctx.destroy()


def run_vocab_ce(rank, world_size, port):
    VOCAB, N = 24, 10
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=world_size)
    torch.manual_seed(11)
    logits = torch.randn(N, VOCAB, requires_grad=True)
    targets = torch.randint(0, VOCAB, (N,))
    loss_ref = TF.cross_entropy(logits, targets)

    local_logits = logits.detach().chunk(world_size, dim=-1)[rank].clone().requires_grad_(True)
    loss_fn = VocabParallelCrossEntropy(parallel_context=ctx)
    loss = loss_fn(local_logits, targets)
    assert torch.allclose(loss, loss_ref, atol=1e-5), (loss, loss_ref)

    loss_ref.backward()
    loss.backward()
    g_ref = logits.grad.chunk(world_size, dim=-1)[rank]
    assert torch.allclose(local_logits.grad, g_ref, atol=1e-5)
    ctx.destroy()


def test_column_parallel_linear():
    spawn(run_column_linear, world_size=2)


def test_row_parallel_linear():
    spawn(run_row_linear, world_size=2)


def test_parallel_embedding():
    spawn(run_embedding, world_size=2)


def test_vocab_parallel_cross_entropy():
    spawn(run_vocab_ce, world_size=2)


def test_column_linear_no_bias_single():
    # reference bug regression: bias=False must work (linear.py:44 there)
    spawn(run_no_bias, world_size=1)


def run_no_bias(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=world_size)
    layer = ColumnParallelLinear(IN, OUT, bias=False, parallel_context=ctx)
    nn.init.normal_(layer.weight)
    y = layer(torch.randn(2, IN))
    assert y.shape == (2, OUT)
    ctx.destroy()
