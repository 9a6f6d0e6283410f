"""FusedAdamW vs torch.optim.AdamW parity.

CPU: the eager fallback path must match torch's AdamW on fp32 params.
GPU: the chunked HIP kernel must match torch on fp32, and the bf16 path
must match an fp32-state reference within bf16 rounding.
"""
import pytest
import torch

from pipegoose_amd.optim.fused_adamw import FusedAdamW


def _models(dtype, device, seed=3):
    torch.manual_seed(seed)
    m1 = torch.nn.Sequential(
        torch.nn.Linear(64, 130), torch.nn.GELU(), torch.nn.Linear(130, 7),
    ).to(device=device, dtype=dtype)
    m2 = torch.nn.Sequential(
        torch.nn.Linear(64, 130), torch.nn.GELU(), torch.nn.Linear(130, 7),
    ).to(device=device, dtype=dtype)
    m2.load_state_dict(m1.state_dict())
    return m1, m2


def _run(model, opt, device, dtype, steps=5, seed=11):
    torch.manual_seed(seed)
    for _ in range(steps):
        x = torch.randn(16, 64, device=device, dtype=dtype)
        loss = model(x).float().pow(2).mean()
        loss.backward()
        opt.step()
        opt.zero_grad(set_to_none=True)


def test_fused_adamw_cpu_matches_torch():
    m1, m2 = _models(torch.float32, "cpu")
    o1 = torch.optim.AdamW(m1.parameters(), lr=1e-2, weight_decay=0.05)
    o2 = FusedAdamW(m2.parameters(), lr=1e-2, weight_decay=0.05)
    _run(m1, o1, "cpu", torch.float32)
    _run(m2, o2, "cpu", torch.float32)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-6), (p1 - p2).abs().max()


def test_fused_adamw_state_dict_roundtrip():
    m1, _ = _models(torch.float32, "cpu")
    opt = FusedAdamW(m1.parameters(), lr=1e-2)
    _run(m1, opt, "cpu", torch.float32, steps=2)
    sd = opt.state_dict()
    # torch AdamW can consume the state dict (same per-param layout)
    ref = torch.optim.AdamW(m1.parameters(), lr=1e-2)
    ref.load_state_dict(sd)
    st = list(ref.state_dict()["state"].values())[0]
    assert "exp_avg" in st and "exp_avg_sq" in st


@pytest.mark.gpu
def test_fused_adamw_gpu_bf16_state_matches_torch():
    """Default state dtype follows the param dtype (torch semantics): bf16
    params -> bf16 moments; compare against torch.optim.AdamW directly."""
    m1, m2 = _models(torch.bfloat16, "cuda")
    o1 = torch.optim.AdamW(m1.parameters(), lr=1e-2, weight_decay=0.05,
                           foreach=True)
    o2 = FusedAdamW(m2.parameters(), lr=1e-2, weight_decay=0.05)
    _run(m1, o1, "cuda", torch.bfloat16)
    _run(m2, o2, "cuda", torch.bfloat16)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1.float(), p2.float(), atol=3e-2), \
            (p1.float() - p2.float()).abs().max()


@pytest.mark.gpu
def test_fused_adamw_gpu_matches_torch_fp32():
    m1, m2 = _models(torch.float32, "cuda")
    o1 = torch.optim.AdamW(m1.parameters(), lr=1e-2, weight_decay=0.05)
    o2 = FusedAdamW(m2.parameters(), lr=1e-2, weight_decay=0.05)
    _run(m1, o1, "cuda", torch.float32)
    _run(m2, o2, "cuda", torch.float32)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.allclose(p1, p2, atol=1e-5), (p1 - p2).abs().max()


@pytest.mark.gpu
def test_fused_adamw_gpu_bf16_vs_fp32_state_reference():
    torch.manual_seed(5)
    # big enough for several slabs + a ragged tail
    shapes = [(300, 257), (65536 + 13,), (31,)]
    ps1 = [torch.randn(s, device="cuda").bfloat16().requires_grad_(True)
           for s in shapes]
    ps2 = [p.detach().clone().requires_grad_(True) for p in ps1]
    o2 = FusedAdamW(ps2, lr=1e-2, weight_decay=0.03,
                    state_dtype=torch.float32)
    # reference: fp32-state AdamW applied manually to bf16 params
    ms = [torch.zeros(s, device="cuda") for s in shapes]
    vs = [torch.zeros(s, device="cuda") for s in shapes]
    b1, b2, eps, lr, wd = 0.9, 0.999, 1e-8, 1e-2, 0.03
    for t in range(1, 5):
        grads = [torch.randn(s, device="cuda").bfloat16() for s in shapes]
        for p, g in zip(ps2, grads):
            p.grad = g.clone()
        o2.step()
        o2.zero_grad()
        for p, g, m, v in zip(ps1, grads, ms, vs):
            gf = g.float()
            m.mul_(b1).add_(gf, alpha=1 - b1)
            v.mul_(b2).addcmul_(gf, gf, value=1 - b2)
            denom = v.sqrt().div_((1 - b2 ** t) ** 0.5).add_(eps)
            upd = p.detach().float() * (1 - lr * wd) - (lr / (1 - b1 ** t)) * m / denom
            p.data.copy_(upd.bfloat16())
    for p1, p2 in zip(ps1, ps2):
        assert torch.allclose(p1.float(), p2.float(), atol=2e-2), \
            (p1.float() - p2.float()).abs().max()
        # states must agree closely (fp32 both sides)
    for p2, m_ref, v_ref in zip(ps2, ms, vs):
        st = o2.state[p2]
        assert torch.allclose(st["exp_avg"], m_ref, atol=1e-4)
        assert torch.allclose(st["exp_avg_sq"], v_ref, atol=1e-5)


def test_fused_adamw_checkpoint_resume(tmp_path):
    """save_training_state/load_training_state round-trips FusedAdamW state
    (weights_only-safe: tensors + primitives only)."""
    import os
    from pipegoose_amd.nn.utils import (load_training_state,
                                        save_training_state)
    from pipegoose_amd.testing.utils import init_parallel_context

    ctx = init_parallel_context(0, 1, 29930)
    m1, m2 = _models(torch.float32, "cpu")
    o1 = FusedAdamW(m1.parameters(), lr=1e-2)
    _run(m1, o1, "cpu", torch.float32, steps=3)
    save_training_state(o1, str(tmp_path), parallel_context=ctx, step=3)

    m2.load_state_dict(m1.state_dict())  # resume = weights + optim state
    o2 = FusedAdamW(m2.parameters(), lr=1e-2)
    payload = load_training_state(o2, str(tmp_path), parallel_context=ctx)
    assert payload["step"] == 3
    # the resumed run must continue EXACTLY like the uninterrupted one
    _run(m2, o2, "cpu", torch.float32, steps=2, seed=13)
    _run(m1, o1, "cpu", torch.float32, steps=2, seed=13)
    for p1, p2 in zip(m1.parameters(), m2.parameters()):
        assert torch.equal(p1, p2)
    ctx.destroy()
