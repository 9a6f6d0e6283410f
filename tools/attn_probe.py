"""Probe sdpa backend selection + timing for various head dims on MI355X."""
import time

import torch
import torch.nn.functional as TF


def bench(fn, iters=10):
    for _ in range(3):
        fn()
    torch.cuda.synchronize()
    t0 = time.perf_counter()
    for _ in range(iters):
        fn()
    torch.cuda.synchronize()
    return (time.perf_counter() - t0) / iters * 1e3


def main():
    B, H, S = 8, 16, 2048
    for D in (64, 66, 72, 128):
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
        k = torch.randn_like(q).requires_grad_(True)
        v = torch.randn_like(q).requires_grad_(True)
        t_fwd = bench(lambda: TF.scaled_dot_product_attention(q, k, v, is_causal=True))
        out = TF.scaled_dot_product_attention(q, k, v, is_causal=True)
        g = torch.randn_like(out)
        t_bwd = bench(lambda: torch.autograd.grad(
            TF.scaled_dot_product_attention(q, k, v, is_causal=True), (q, k, v), g))
        print(f"D={D}: fwd {t_fwd:.2f} ms  fwd+bwd {t_bwd:.2f} ms", flush=True)

    # with additive mask (the old path)
    D = 64
    q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16, requires_grad=True)
    k, v = torch.randn_like(q), torch.randn_like(q)
    mask = torch.randn(1, H, S, S, device="cuda", dtype=torch.bfloat16)
    t_fwd = bench(lambda: TF.scaled_dot_product_attention(q, k, v, attn_mask=mask))
    print(f"D=64+mask: fwd {t_fwd:.2f} ms", flush=True)

    # which backends can run the folded shape?
    from torch.nn.attention import SDPBackend, sdpa_kernel
    for D in (64, 66, 72):
        q = torch.randn(B, H, S, D, device="cuda", dtype=torch.bfloat16)
        k, v = torch.randn_like(q), torch.randn_like(q)
        for be in (SDPBackend.FLASH_ATTENTION, SDPBackend.EFFICIENT_ATTENTION):
            try:
                with sdpa_kernel(be):
                    TF.scaled_dot_product_attention(q, k, v, is_causal=True)
                ok = "ok"
            except Exception as e:
                ok = f"FAIL {type(e).__name__}"
            print(f"D={D} {be.name}: {ok}", flush=True)


if __name__ == "__main__":
    main()
