"""fp8 linear path (ops/fp8.py): quantization, surgery, GEMM parity.

CPU tests cover the quantizer and the class-swap surgery (the fp8 GEMM
itself needs hipBLASLt, so numerics run under @pytest.mark.gpu against a
bf16 nn.Linear oracle)."""
import pytest
import torch
import torch.nn as nn

from pipegoose_amd.ops.fp8 import (Fp8Linear, convert_linear_to_fp8,
                                   dequantize, quantize, revert_fp8)


def test_quantize_roundtrip_cpu():
    torch.manual_seed(0)
    t = torch.randn(256, 128) * 3.0
    for dtype, tol in [(torch.float8_e4m3fn, 0.05), (torch.float8_e5m2, 0.1)]:
        q, s = quantize(t, dtype)
        back = dequantize(q, s)
        rel = (back - t).abs().mean() / t.abs().mean()
        assert rel < tol, f"{dtype}: relerr {rel:.3f}"
    # scale covers the actual amax: no inf/nan in the quantized tensor
    q, _ = quantize(t * 1e4, torch.float8_e4m3fn)
    assert torch.isfinite(q.float()).all()


def test_convert_and_revert():
    m = nn.Sequential(nn.Linear(32, 64), nn.ReLU(), nn.Linear(64, 32))
    assert convert_linear_to_fp8(m) == 2
    assert type(m[0]) is Fp8Linear and type(m[2]) is Fp8Linear
    assert revert_fp8(m) == 2
    assert type(m[0]) is nn.Linear

    m = nn.ModuleDict({"dense_h_to_4h": nn.Linear(8, 16),
                       "proj": nn.Linear(16, 8)})
    assert convert_linear_to_fp8(m, names=["dense_h_to_4h"]) == 1
    assert type(m["proj"]) is nn.Linear


def test_fp8_linear_cpu_fallback():
    torch.manual_seed(0)
    lin = nn.Linear(32, 48)
    x = torch.randn(4, 32)
    want = lin(x)
    lin.__class__ = Fp8Linear
    assert torch.equal(lin(x), want)   # CPU path is plain F.linear


@pytest.mark.gpu
def test_fp8_linear_matches_bf16_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    torch.manual_seed(0)
    M, K, N = 256, 512, 384
    lin = nn.Linear(K, N).to("cuda", torch.bfloat16)
    x = torch.randn(M, K, device="cuda", dtype=torch.bfloat16,
                    requires_grad=True)
    y_ref = lin(x)
    g = torch.randn_like(y_ref)
    y_ref.backward(g)
    gx_ref, gw_ref, gb_ref = (x.grad.clone(), lin.weight.grad.clone(),
                              lin.bias.grad.clone())

    x.grad = None
    lin.zero_grad()
    lin.__class__ = Fp8Linear
    y = lin(x)
    y.backward(g)

    def rel(a, b):
        return ((a.float() - b.float()).abs().mean()
                / b.float().abs().mean()).item()

    assert rel(y, y_ref) < 0.08, f"fwd relerr {rel(y, y_ref)}"
    assert rel(x.grad, gx_ref) < 0.15
    assert rel(lin.weight.grad, gw_ref) < 0.15
    assert rel(lin.bias.grad, gb_ref) < 0.02   # bias grad stays bf16-exact


def test_tp_layer_fp8_flag_cpu_fallback():
    """convert_linear_to_fp8 flags Column/RowParallelLinear; off-GPU the
    flagged forward must fall back to the exact bf16/fp32 path."""
    import os
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29514")
    from pipegoose_amd.testing.utils import init_parallel_context
    from pipegoose_amd.nn.tensor_parallel.linear import (ColumnParallelLinear,
                                                         RowParallelLinear)
    ctx = init_parallel_context(0, 1, 29514)
    try:
        torch.manual_seed(0)
        col = ColumnParallelLinear(32, 64, parallel_context=ctx)
        row = RowParallelLinear(64, 32, parallel_context=ctx)
        for m in (col, row):
            torch.nn.init.normal_(m.weight)
            torch.nn.init.normal_(m.bias)
        x = torch.randn(4, 32)
        want = row(col(x))
        holder = nn.ModuleDict({"c": col, "r": row})
        assert convert_linear_to_fp8(holder) == 2
        assert col.fp8 and row.fp8
        assert torch.equal(row(col(x)), want)
        assert revert_fp8(holder) == 2
    finally:
        ctx.destroy()


@pytest.mark.gpu
def test_fp8_quant_kernel_matches_eager_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from pipegoose_amd.ops import get_extension
    from pipegoose_amd.ops.fp8 import E4M3_MAX, E5M2_MAX
    ext = get_extension(required=True)
    torch.manual_seed(0)
    for n, dtype, e5m2, maxv in [((4096, 1024), torch.float8_e4m3fn, False,
                                  E4M3_MAX),
                                 ((333,), torch.float8_e5m2, True, E5M2_MAX)]:
        t = torch.randn(*n, device="cuda", dtype=torch.bfloat16) * 5.0
        q, sc = ext.fp8_quant(t, e5m2)
        assert q.dtype == dtype and q.shape == t.shape
        # eager oracle (same math): amax, scale, clamp, RNE cast
        amax = t.abs().amax().float().clamp(min=1e-12)
        want_scale = amax / maxv
        assert torch.allclose(sc[0], want_scale, rtol=1e-6)
        want = (t.float() / want_scale).clamp(-maxv, maxv).to(dtype)
        mism = (q.view(torch.uint8) != want.view(torch.uint8)).sum().item()
        assert mism <= q.numel() * 1e-3, f"{mism} byte mismatches"
        # dequantized roundtrip sanity
        back = q.float() * sc[0]
        rel = (back - t.float()).abs().mean() / t.float().abs().mean()
        assert rel < (0.1 if e5m2 else 0.05)


@pytest.mark.gpu
def test_fp8_quant_dual_matches_separate_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from pipegoose_amd.ops import get_extension
    ext = get_extension(required=True)
    torch.manual_seed(3)
    for (r, c), e5m2 in [((512, 256), False), ((48, 132), True)]:
        t = torch.randn(r, c, device="cuda", dtype=torch.bfloat16) * 2.0
        q, qt, sc = ext.fp8_quant_dual(t, e5m2)
        q_ref, sc_ref = ext.fp8_quant(t, e5m2)
        assert torch.allclose(sc, sc_ref)
        assert torch.equal(q.view(torch.uint8), q_ref.view(torch.uint8))
        assert torch.equal(qt.view(torch.uint8),
                           q_ref.view(torch.uint8).t().contiguous())


@pytest.mark.gpu
def test_fp8_transpose_kernel_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    from pipegoose_amd.ops import get_extension
    ext = get_extension(required=True)
    for (r, c) in [(128, 256), (65, 130), (4096, 1024)]:
        t = torch.randint(0, 255, (r, c), device="cuda",
                          dtype=torch.uint8).view(torch.float8_e4m3fn)
        out = ext.fp8_transpose(t)
        assert out.shape == (c, r)
        assert torch.equal(out.view(torch.uint8),
                           t.view(torch.uint8).t().contiguous())


@pytest.mark.gpu
def test_fp8_bloom_mlp_forward_gpu():
    if not torch.cuda.is_available():
        pytest.skip("no GPU")
    import os
    os.environ.setdefault("RANK", "0")
    os.environ.setdefault("LOCAL_RANK", "0")
    os.environ.setdefault("WORLD_SIZE", "1")
    os.environ.setdefault("MASTER_ADDR", "127.0.0.1")
    os.environ.setdefault("MASTER_PORT", "29531")
    from pipegoose_amd.distributed.parallel_context import ParallelContext
    from pipegoose_amd.models.bloom import BloomForCausalLM, bloom_tiny

    ctx = ParallelContext.get_context() or ParallelContext.from_torch()
    torch.manual_seed(0)
    model = BloomForCausalLM(bloom_tiny(), ctx).to("cuda", torch.bfloat16)
    ids = torch.randint(0, 100, (16, 32), device="cuda")
    with torch.no_grad():
        ref = model(ids)
    n = convert_linear_to_fp8(model, names=["dense_h_to_4h", "dense_4h_to_h"])
    assert n > 0
    with torch.no_grad():
        out = model(ids)
    rel = ((out.float() - ref.float()).abs().mean()
           / ref.float().abs().mean()).item()
    assert rel < 0.1, f"MLP-fp8 logits relerr {rel}"
