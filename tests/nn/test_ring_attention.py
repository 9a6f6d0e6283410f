"""Context-parallel (ring-style) attention vs full-sequence oracle."""
import math

import torch

from pipegoose_amd.nn.ring_attention import (_block_attention_ref,
                                             ring_attention)
from pipegoose_amd.testing.utils import init_parallel_context, spawn


def _full_oracle(q, k, v, slopes, scale):
    o, _ = _block_attention_ref(q, k, v, slopes, scale, 0)
    return o


def test_block_attention_offset_semantics():
    """kv_off <= -S means fully visible; 0 means causal-diagonal."""
    torch.manual_seed(70)
    B, H, S, D = 1, 2, 8, 16
    q = torch.randn(B, H, S, D)
    k, v = torch.randn_like(q), torch.randn_like(q)
    slopes = torch.rand(H) * 0.3
    # full-visibility block == attention with no mask
    o_full, _ = _block_attention_ref(q, k, v, slopes, 0.25, -S)
    jk = torch.arange(S) - S
    rel = jk[None, :] - torch.arange(S)[:, None]
    bias = slopes[:, None, None] * rel[None].float()
    scores = (q.float() @ k.float().transpose(-1, -2)) * 0.25 + bias[None]
    ref = torch.softmax(scores, -1) @ v.float()
    assert torch.allclose(o_full, ref, atol=1e-6)


def _run_ring_cp2(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=2)
    torch.manual_seed(71)  # same full tensors everywhere
    B, H, S, D = 2, 3, 16, 32
    qf = torch.randn(B, H, S, D)
    kf = torch.randn_like(qf)
    vf = torch.randn_like(qf)
    slopes = torch.rand(H) * 0.4
    scale = 1.0 / math.sqrt(D)

    ref = _full_oracle(qf, kf, vf, slopes, scale)

    Sl = S // 2
    sl = slice(rank * Sl, (rank + 1) * Sl)
    q = qf[:, :, sl].clone().requires_grad_(True)
    k = kf[:, :, sl].clone().requires_grad_(True)
    v = vf[:, :, sl].clone().requires_grad_(True)
    out = ring_attention(q, k, v, slopes, scale, parallel_context=ctx)
    assert torch.allclose(out.float(), ref[:, :, sl], atol=1e-5), \
        (out.float() - ref[:, :, sl]).abs().max()

    # grads: compare against the full-sequence autograd oracle
    torch.manual_seed(72)
    g_full = torch.randn_like(ref)
    out.backward(g_full[:, :, sl])

    qf2 = qf.clone().requires_grad_(True)
    kf2 = kf.clone().requires_grad_(True)
    vf2 = vf.clone().requires_grad_(True)
    _full_oracle(qf2, kf2, vf2, slopes, scale).backward(g_full)
    assert torch.allclose(q.grad, qf2.grad[:, :, sl], atol=1e-5)
    assert torch.allclose(k.grad, kf2.grad[:, :, sl], atol=1e-5), \
        (k.grad - kf2.grad[:, :, sl]).abs().max()
    assert torch.allclose(v.grad, vf2.grad[:, :, sl], atol=1e-5)
    ctx.destroy()


def test_ring_attention_cp2_matches_full():
    spawn(_run_ring_cp2, world_size=2)


def _run_ring_cp4(rank, world_size, port):
    ctx = init_parallel_context(rank, world_size, port, tensor_parallel_size=4)
    torch.manual_seed(73)
    B, H, S, D = 1, 2, 32, 16
    qf = torch.randn(B, H, S, D)
    kf, vf = torch.randn_like(qf), torch.randn_like(qf)
    slopes = torch.zeros(H)  # plain causal (llama-style)
    scale = 0.25
    ref = _full_oracle(qf, kf, vf, slopes, scale)
    Sl = S // 4
    sl = slice(rank * Sl, (rank + 1) * Sl)
    out = ring_attention(qf[:, :, sl], kf[:, :, sl], vf[:, :, sl],
                         slopes, scale, parallel_context=ctx)
    assert torch.allclose(out.float(), ref[:, :, sl], atol=1e-5)
    ctx.destroy()


def test_ring_attention_cp4_matches_full():
    spawn(_run_ring_cp4, world_size=4)


# ------------------------- rotation-based (memory-bounded) CP over CONTEXT

def _run_rotate(rank, world_size, port, with_alibi):
    ctx = init_parallel_context(rank, world_size, port,
                                context_parallel_size=world_size)
    from pipegoose_amd.distributed.parallel_mode import ParallelMode
    from pipegoose_amd.nn.ring_attention import ring_attention_rotate
    torch.manual_seed(11)
    B, H, S, D = 2, 3, 16 * world_size, 8
    qf = torch.randn(B, H, S, D)
    kf = torch.randn(B, H, S, D)
    vf = torch.randn(B, H, S, D)
    slopes = (torch.rand(H) * 0.3) if with_alibi else torch.zeros(H)
    scale = D ** -0.5

    # full-sequence fp32 oracle with grads
    q0 = qf.clone().requires_grad_(True)
    k0 = kf.clone().requires_grad_(True)
    v0 = vf.clone().requires_grad_(True)
    ref, _ = _block_attention_ref(q0, k0, v0, slopes, scale, 0)
    g = torch.randn_like(ref)
    ref.backward(g)

    Sl = S // world_size
    sl = slice(rank * Sl, (rank + 1) * Sl)
    q = qf[:, :, sl].clone().requires_grad_(True)
    k = kf[:, :, sl].clone().requires_grad_(True)
    v = vf[:, :, sl].clone().requires_grad_(True)
    out = ring_attention_rotate(q, k, v, slopes, scale, parallel_context=ctx,
                                parallel_mode=ParallelMode.CONTEXT)
    assert torch.allclose(out, ref[:, :, sl].to(out.dtype), atol=1e-4), \
        (out - ref[:, :, sl]).abs().max()
    out.backward(g[:, :, sl].to(out.dtype))
    for got, want, name in ((q.grad, q0.grad[:, :, sl], "dq"),
                            (k.grad, k0.grad[:, :, sl], "dk"),
                            (v.grad, v0.grad[:, :, sl], "dv")):
        assert torch.allclose(got, want.to(got.dtype), atol=1e-4), \
            (name, (got - want).abs().max())
    ctx.destroy()


def test_ring_rotate_cp2_fwd_bwd_parity():
    spawn(_run_rotate, world_size=2, with_alibi=True)


def test_ring_rotate_cp4_fwd_bwd_parity():
    spawn(_run_rotate, world_size=4, with_alibi=False)
