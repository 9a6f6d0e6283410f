"""Fused AdamW: the whole decoupled-AdamW update in ONE HIP kernel per step.

Drop-in for ``torch.optim.AdamW`` (same hyperparameters, same per-param
``exp_avg``/``exp_avg_sq`` state_dict layout, fp32 state).  On CPU or when
the extension is unavailable it falls back to eager math with identical
numerics, so the class is usable everywhere (tests run it on CPU).

Why: torch's foreach AdamW issues several multi-tensor elementwise kernels
per step, each re-reading the fp32 moments from HBM; the fused kernel does
one read-modify-write pass (p,g in param dtype; m,v fp32).  The r1 profile
put the optimizer at ~7% of the BLOOM-7B1 step (ROADMAP.md §3).

No reference counterpart: the reference ran plain torch.optim under ZeRO-1
(optim/zero/optim.py:23-33).
"""
from typing import List

import torch

from pipegoose_amd.ops import get_extension

_SLAB = 65536


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self._meta_cache = None  # (key, meta_dev, n_slabs, lists)

    def _build_meta(self, params: List[torch.Tensor],
                    grads: List[torch.Tensor],
                    ms: List[torch.Tensor], vs: List[torch.Tensor],
                    device) -> tuple:
        T = len(params)
        ptrs = torch.empty(4 * T, dtype=torch.int64)
        numels = torch.empty(T, dtype=torch.int64)
        st, si = [], []
        for i, (p, g, m, v) in enumerate(zip(params, grads, ms, vs)):
            ptrs[i] = p.data_ptr()
            ptrs[T + i] = g.data_ptr()
            ptrs[2 * T + i] = m.data_ptr()
            ptrs[3 * T + i] = v.data_ptr()
            numels[i] = p.numel()
            for s in range((p.numel() + _SLAB - 1) // _SLAB):
                st.append(i)
                si.append(s)
        n_slabs = len(st)
        meta = torch.cat([
            ptrs.view(torch.uint8),
            numels.view(torch.uint8),
            torch.tensor(st, dtype=torch.int32).view(torch.uint8),
            torch.tensor(si, dtype=torch.int32).view(torch.uint8),
        ])
        return meta.to(device, non_blocking=True), n_slabs

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        ext = get_extension()
        for group in self.param_groups:
            params, grads, ms, vs = [], [], [], []
            step_t = None
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    state["step"] = torch.zeros((), dtype=torch.float32)
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=torch.float32, memory_format=torch.contiguous_format)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=torch.float32, memory_format=torch.contiguous_format)
                state["step"] += 1
                step_t = int(state["step"].item())
                params.append(p)
                grads.append(p.grad)
                ms.append(state["exp_avg"])
                vs.append(state["exp_avg_sq"])
            if not params:
                continue
            lr = group["lr"]
            beta1, beta2 = group["betas"]
            eps, wd = group["eps"], group["weight_decay"]
            use_kernel = (
                ext is not None and hasattr(ext, "adamw_fused_step")
                and params[0].is_cuda
                and params[0].dtype in (torch.bfloat16, torch.float32)
                and all(p.is_contiguous() and g.is_contiguous()
                        and g.dtype == p.dtype and p.dtype == params[0].dtype
                        for p, g in zip(params, grads))
            )
            if use_kernel:
                key = tuple(g.data_ptr() for g in grads) + \
                    tuple(p.data_ptr() for p in params)
                if self._meta_cache is None or self._meta_cache[0] != key:
                    meta, n_slabs = self._build_meta(
                        params, grads, ms, vs, params[0].device)
                    self._meta_cache = (key, meta, n_slabs)
                _, meta, n_slabs = self._meta_cache
                ext.adamw_fused_step(params, grads, ms, vs, meta, n_slabs,
                                     lr, beta1, beta2, eps, wd, step_t)
            else:
                for p, g, m, v in zip(params, grads, ms, vs):
                    t = int(self.state[p]["step"].item())
                    gf = g.float()
                    m.mul_(beta1).add_(gf, alpha=1 - beta1)
                    v.mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
                    bc1 = 1 - beta1 ** t
                    bc2 = 1 - beta2 ** t
                    denom = v.sqrt().div_(bc2 ** 0.5).add_(eps)
                    upd = p.float() * (1 - lr * wd) - (lr / bc1) * m / denom
                    p.copy_(upd.to(p.dtype))
        return loss
