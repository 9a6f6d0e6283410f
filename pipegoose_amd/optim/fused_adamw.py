"""Fused AdamW: the whole decoupled-AdamW update in ONE HIP kernel per step.

Drop-in for ``torch.optim.AdamW`` (same hyperparameters, same per-param
``exp_avg``/``exp_avg_sq`` state_dict layout; state dtype follows the param
dtype like torch's, or fp32 via ``state_dtype=torch.float32``).  On CPU or
when the extension is unavailable it falls back to eager math with the same
numerics, so the class is usable everywhere (tests run it on CPU).

Why: torch's foreach AdamW issues several multi-tensor elementwise kernels
per step, each re-reading the moments from HBM; the fused kernel does one
read-modify-write pass.  The r1 profile put the optimizer at ~7% of the
BLOOM-7B1 step (ROADMAP.md §3).

The slab table (which workgroup updates which 64K-element slice of which
tensor) depends only on the tensor SIZES, so it is built once and uploaded
once; only the small 4*T pointer array refreshes when grad pointers change
(``zero_grad(set_to_none=True)`` reallocates grads every step).

No reference counterpart: the reference ran plain torch.optim under ZeRO-1
(optim/zero/optim.py:23-33).
"""
from typing import List, Optional

import torch

from pipegoose_amd.ops import get_extension

_SLAB = 65536


class FusedAdamW(torch.optim.Optimizer):
    def __init__(self, params, lr=1e-3, betas=(0.9, 0.999), eps=1e-8,
                 weight_decay=1e-2, state_dtype: Optional[torch.dtype] = None):
        defaults = dict(lr=lr, betas=betas, eps=eps,
                        weight_decay=weight_decay)
        super().__init__(params, defaults)
        self.state_dtype = state_dtype  # None = follow param dtype (torch)
        # per-param-group caches (a single slot would thrash when several
        # groups alternate within one step)
        self._static_meta = {}  # group idx -> (shape_key, tail, n_slabs)
        self._ptr_meta = {}     # group idx -> (ptr_key, meta_dev)

    def _meta(self, gi: int, params: List[torch.Tensor],
              grads: List[torch.Tensor], ms: List[torch.Tensor],
              vs: List[torch.Tensor], device):
        T = len(params)
        shape_key = (T, tuple(p.numel() for p in params))
        if (gi not in self._static_meta
                or self._static_meta[gi][0] != shape_key):
            numels = torch.tensor([p.numel() for p in params],
                                  dtype=torch.int64)
            st, si = [], []
            for i, p in enumerate(params):
                n_slab = (p.numel() + _SLAB - 1) // _SLAB
                st.extend([i] * n_slab)
                si.extend(range(n_slab))
            tail = torch.cat([
                numels.view(torch.uint8),
                torch.tensor(st, dtype=torch.int32).view(torch.uint8),
                torch.tensor(si, dtype=torch.int32).view(torch.uint8),
            ])
            self._static_meta[gi] = (shape_key, tail, len(st))
            self._ptr_meta.pop(gi, None)
        _, tail, n_slabs = self._static_meta[gi]
        # key over grads AND params: grads reallocate every step under
        # set_to_none=True, while params can move under a ZeRO flat-shard
        # rebuild with grads untouched (set_to_none=False)
        ptr_key = tuple(t.data_ptr() for t in grads) + \
            tuple(p.data_ptr() for p in params)
        if gi not in self._ptr_meta or self._ptr_meta[gi][0] != ptr_key:
            ptrs = torch.empty(4 * T, dtype=torch.int64)
            for i, (p, g, m, v) in enumerate(zip(params, grads, ms, vs)):
                ptrs[i] = p.data_ptr()
                ptrs[T + i] = g.data_ptr()
                ptrs[2 * T + i] = m.data_ptr()
                ptrs[3 * T + i] = v.data_ptr()
            meta = torch.cat([ptrs.view(torch.uint8), tail]).to(
                device, non_blocking=True)
            self._ptr_meta[gi] = (ptr_key, meta)
        return self._ptr_meta[gi][1], n_slabs

    @torch.no_grad()
    def step(self, closure=None):
        loss = None
        if closure is not None:
            with torch.enable_grad():
                loss = closure()
        ext = get_extension()
        for gi, group in enumerate(self.param_groups):
            params, grads, ms, vs = [], [], [], []
            step_t = None
            for p in group["params"]:
                if p.grad is None:
                    continue
                state = self.state[p]
                if len(state) == 0:
                    sdt = self.state_dtype or p.dtype
                    state["step"] = torch.zeros((), dtype=torch.float32)
                    state["exp_avg"] = torch.zeros_like(
                        p, dtype=sdt, memory_format=torch.contiguous_format)
                    state["exp_avg_sq"] = torch.zeros_like(
                        p, dtype=sdt, memory_format=torch.contiguous_format)
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
            sdt0 = ms[0].dtype
            use_kernel = (
                ext is not None and hasattr(ext, "adamw_fused_step")
                and params[0].is_cuda
                and (params[0].dtype, sdt0) in (
                    (torch.bfloat16, torch.float32),
                    (torch.bfloat16, torch.bfloat16),
                    (torch.float32, torch.float32))
                and all(p.is_contiguous() and g.is_contiguous()
                        and g.dtype == p.dtype and p.dtype == params[0].dtype
                        and m.dtype == sdt0
                        for p, g, m in zip(params, grads, ms))
            )
            if use_kernel:
                meta, n_slabs = self._meta(gi, params, grads, ms, vs,
                                           params[0].device)
                ext.adamw_fused_step(params, grads, ms, vs, meta, n_slabs,
                                     lr, beta1, beta2, eps, wd, step_t)
            else:
                for p, g, m, v in zip(params, grads, ms, vs):
                    t = int(self.state[p]["step"].item())
                    gf = g.float()
                    mf = m.float().mul_(beta1).add_(gf, alpha=1 - beta1)
                    vf = v.float().mul_(beta2).addcmul_(gf, gf, value=1 - beta2)
                    m.copy_(mf.to(m.dtype))
                    v.copy_(vf.to(v.dtype))
                    bc1 = 1 - beta1 ** t
                    bc2 = 1 - beta2 ** t
                    denom = vf.sqrt().div_(bc2 ** 0.5).add_(eps)
                    upd = p.float() * (1 - lr * wd) - (lr / bc1) * mf / denom
                    p.copy_(upd.to(p.dtype))
        return loss
