"""Blockwise context-parallel ("ring") attention over xGMI.

Long sequences are sharded along S across a process group (rank order =
sequence order).  Each rank computes exact causal(+ALiBi) attention for its
Q shard against every visible KV shard blockwise, merging the per-block
outputs with log-sum-exp weights — the flash-attention merge — so the
result is bit-for-bit the attention of the full sequence.

KV movement rides `all_gather_sequence` (fwd all-gather / bwd reduce-scatter
along S): RCCL implements the all-gather as a ring over the xGMI links, so
the wire pattern matches classic ring attention; the v1 trade-off is that
each rank holds the gathered KV during the block loop (no rotation-step
memory bound) — activations and the O(S²) attention work ARE divided by cp.

Absent in the reference (README listed "sequence parallelism" but shipped
none — SURVEY.md §5 'Long-context').
"""
import math
from typing import Optional

import torch

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.tensor_parallel._functional import all_gather_sequence
from pipegoose_amd.ops import get_extension


def _block_attention_ref(q, k, v, slopes, scale, kv_off):
    """fp32 torch oracle returning (normalized o, lse). kv positions are
    globally shifted by kv_off relative to q positions."""
    B, H, Sq, D = q.shape
    Sk = k.size(2)
    iq = torch.arange(Sq, device=q.device)
    jk = torch.arange(Sk, device=q.device) + kv_off
    rel = jk[None, :] - iq[:, None]                       # j_global - i
    bias = slopes.float().to(q.device)[:, None, None] * rel[None].float()
    bias = bias.masked_fill(rel[None] > 0, float("-inf"))
    scores = (q.float() @ k.float().transpose(-1, -2)) * scale + bias[None]
    lse = torch.logsumexp(scores, dim=-1)                 # [B,H,Sq]
    o = torch.softmax(scores, dim=-1) @ v.float()
    return o, lse


class _BlockAttn(torch.autograd.Function):
    """GPU block attention with kv position offset (hand-written kernels)."""

    @staticmethod
    def forward(ctx, q, k, v, slopes, scale, kv_off):
        ext = get_extension(required=True)
        o, lse = ext.attn_fwd(q, k, v, slopes, scale, kv_off)
        ctx.save_for_backward(q, k, v, o, lse, slopes)
        ctx.scale, ctx.kv_off = scale, kv_off
        return o.float(), lse

    @staticmethod
    def backward(ctx, do, dlse):
        ext = get_extension(required=True)
        q, k, v, o, lse, slopes = ctx.saved_tensors
        do = do.to(q.dtype)
        do = do if do.stride(-1) == 1 else do.contiguous()
        dq, dk, dv = ext.attn_bwd(do, q, k, v, o, lse, slopes,
                                  ctx.scale, ctx.kv_off)
        return dq, dk, dv, None, None, None


def _block_attention(q, k, v, slopes, scale, kv_off):
    use_kernel = (q.is_cuda and q.dtype == torch.bfloat16
                  and q.size(-1) in (64, 128) and q.size(-2) % 64 == 0
                  and get_extension() is not None)
    if use_kernel:
        return _BlockAttn.apply(q.contiguous(), k.contiguous(),
                                v.contiguous(),
                                slopes.to(q.device, torch.float32),
                                scale, kv_off)
    return _block_attention_ref(q, k, v, slopes, scale, kv_off)


def _ring_exchange(t: torch.Tensor, pc: ParallelContext,
                   mode: ParallelMode, reverse: bool = False) -> torch.Tensor:
    """One rotation step: send to the next rank on the ring, receive from the
    previous (reversed for the backward cycle).  Same call count/order on
    every rank, so the matches can't deadlock."""
    import torch.distributed as dist
    group = pc.get_group(mode)
    nxt = pc.get_next_global_rank(mode)
    prv = pc.get_prev_global_rank(mode)
    dst, src = ((nxt, prv) if not reverse else (prv, nxt))
    buf = torch.empty_like(t)
    ops = [dist.P2POp(dist.isend, t.contiguous(), dst, group),
           dist.P2POp(dist.irecv, buf, src, group)]
    for w in dist.batch_isend_irecv(ops):
        w.wait()
    return buf


def _block_attention_bwd_ref(do, q, k, v, o, lse, slopes, scale, kv_off):
    """fp32 oracle blockwise backward from the GLOBAL lse (flash identity:
    p_ij = exp(s_ij - lse_i) is exact per block)."""
    B, H, Sq, D = q.shape
    Sk = k.size(2)
    iq = torch.arange(Sq, device=q.device)
    jk = torch.arange(Sk, device=q.device) + kv_off
    rel = jk[None, :] - iq[:, None]
    bias = slopes.float().to(q.device)[:, None, None] * rel[None].float()
    bias = bias.masked_fill(rel[None] > 0, float("-inf"))
    s = (q.float() @ k.float().transpose(-1, -2)) * scale + bias[None]
    p = torch.exp(s - lse[..., None])
    dv = p.transpose(-1, -2) @ do.float()
    dp = do.float() @ v.float().transpose(-1, -2)
    delta = (do.float() * o.float()).sum(-1, keepdim=True)
    ds = p * (dp - delta) * scale
    dq = ds @ k.float()
    dk = ds.transpose(-1, -2) @ q.float()
    return dq, dk, dv


class _RingAttentionRotate(torch.autograd.Function):
    """Rotation-based context-parallel attention with BOUNDED memory.

    Forward rotates the KV pair around the CONTEXT ring (cp-1 exchanges),
    merging blockwise flash outputs by log-sum-exp; only the LOCAL q/k/v and
    the final (o, lse) are saved — received blocks are never retained, so
    per-rank memory is O(S/cp) regardless of cp (the r1 gathered-KV version
    held the full KV).  Backward re-rotates: a full cycle of cp exchanges
    carries (k, v, dk_acc, dv_acc) so every owner's dk/dv arrives home fully
    accumulated; blockwise grads come from the kv_off backward kernels using
    the GLOBAL lse (exact by the flash identity).
    """

    @staticmethod
    def forward(ctx, q, k, v, slopes, scale, pc, mode):
        cp = pc.get_world_size(mode)
        rank = pc.get_local_rank(mode)
        S_local = q.size(2)
        o_num = w_sum = m_run = None
        cur_k, cur_v = k, v
        with torch.no_grad():
            for step in range(cp):
                owner = (rank - step) % cp
                if owner <= rank:
                    kv_off = (owner - rank) * S_local
                    o_r, lse_r = _block_attention(q, cur_k.contiguous(),
                                                  cur_v.contiguous(),
                                                  slopes, scale, kv_off)
                    lse_r = lse_r[..., None]
                    if o_num is None:
                        m_run, o_num = lse_r, o_r
                        w_sum = torch.ones_like(lse_r)
                    else:
                        m_new = torch.maximum(m_run, lse_r)
                        o_num = o_num * torch.exp(m_run - m_new) \
                            + o_r * torch.exp(lse_r - m_new)
                        w_sum = w_sum * torch.exp(m_run - m_new) \
                            + torch.exp(lse_r - m_new)
                        m_run = m_new
                if step + 1 < cp:
                    cur_k = _ring_exchange(cur_k, pc, mode)
                    cur_v = _ring_exchange(cur_v, pc, mode)
        o = (o_num / w_sum).to(q.dtype)
        lse = (m_run + torch.log(w_sum)).squeeze(-1)  # [B,H,S] global
        ctx.save_for_backward(q, k, v, slopes, o, lse)
        ctx.scale, ctx.pc, ctx.mode = scale, pc, mode
        return o

    @staticmethod
    def backward(ctx, do):
        q, k, v, slopes, o, lse = ctx.saved_tensors
        pc, mode, scale = ctx.pc, ctx.mode, ctx.scale
        cp = pc.get_world_size(mode)
        rank = pc.get_local_rank(mode)
        S_local = q.size(2)
        do = do.contiguous()

        use_kernel = (q.is_cuda and q.dtype == torch.bfloat16
                      and q.size(-1) in (64, 128) and S_local % 64 == 0
                      and get_extension() is not None)
        dq = torch.zeros_like(q, dtype=torch.float32)
        cur_k, cur_v = k, v
        dk_acc = torch.zeros_like(k, dtype=torch.float32)
        dv_acc = torch.zeros_like(v, dtype=torch.float32)
        for step in range(cp):
            owner = (rank - step) % cp
            if owner <= rank:
                kv_off = (owner - rank) * S_local
                if use_kernel:
                    ext = get_extension(required=True)
                    dq_c, dk_c, dv_c = ext.attn_bwd(
                        do.to(q.dtype), q, cur_k.contiguous(),
                        cur_v.contiguous(), o, lse.contiguous(),
                        slopes, scale, kv_off)
                else:
                    dq_c, dk_c, dv_c = _block_attention_bwd_ref(
                        do, q, cur_k, cur_v, o, lse, slopes, scale, kv_off)
                dq += dq_c.float()
                dk_acc += dk_c.float()
                dv_acc += dv_c.float()
            # rotate a full cycle so every dk/dv lands back on its owner
            cur_k = _ring_exchange(cur_k, pc, mode)
            cur_v = _ring_exchange(cur_v, pc, mode)
            dk_acc = _ring_exchange(dk_acc, pc, mode)
            dv_acc = _ring_exchange(dv_acc, pc, mode)
        return (dq.to(q.dtype), dk_acc.to(k.dtype), dv_acc.to(v.dtype),
                None, None, None, None)


def ring_attention_rotate(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    slopes: Optional[torch.Tensor] = None,
    scale: Optional[float] = None,
    parallel_context: ParallelContext = None,
    parallel_mode: ParallelMode = ParallelMode.CONTEXT,
) -> torch.Tensor:
    """Rotation-based (memory-bounded) context-parallel attention over the
    CONTEXT ring; see _RingAttentionRotate.  q/k/v: this rank's sequence
    shard [B, H, S_local, D]."""
    cp = parallel_context.get_world_size(parallel_mode)
    H = q.size(1)
    if slopes is None:
        slopes = torch.zeros(H, device=q.device)
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))
    if cp == 1:
        o, _ = _block_attention(q, k, v, slopes, scale, 0)
        return o.to(q.dtype)
    return _RingAttentionRotate.apply(q, k, v, slopes, scale,
                                      parallel_context, parallel_mode)


def ring_attention(
    q: torch.Tensor, k: torch.Tensor, v: torch.Tensor,
    slopes: Optional[torch.Tensor] = None,
    scale: Optional[float] = None,
    parallel_context: ParallelContext = None,
    parallel_mode: ParallelMode = ParallelMode.TENSOR,
) -> torch.Tensor:
    """q/k/v: this rank's sequence shard [B, H, S_local, D] (rank order ==
    sequence order over the group).  Returns this rank's output shard,
    numerically equal to full-sequence causal(+ALiBi) attention."""
    cp = parallel_context.get_world_size(parallel_mode)
    rank = parallel_context.get_local_rank(parallel_mode)
    H = q.size(1)
    S_local = q.size(2)
    if slopes is None:
        slopes = torch.zeros(H, device=q.device)
    if scale is None:
        scale = 1.0 / math.sqrt(q.size(-1))

    if cp == 1:
        o, _ = _block_attention(q, k, v, slopes, scale, 0)
        return o.to(q.dtype)

    k_full = all_gather_sequence(k, parallel_context, dim=2)
    v_full = all_gather_sequence(v, parallel_context, dim=2)

    o_num = None
    w_sum = None
    m_run = None
    for owner in range(rank + 1):  # causal: only owners at or before us
        k_blk = k_full[:, :, owner * S_local:(owner + 1) * S_local]
        v_blk = v_full[:, :, owner * S_local:(owner + 1) * S_local]
        kv_off = (owner - rank) * S_local
        o_r, lse_r = _block_attention(q, k_blk, v_blk, slopes, scale, kv_off)
        lse_r = lse_r[..., None]                      # [B,H,S,1]
        if o_num is None:
            m_run = lse_r
            o_num = o_r
            w_sum = torch.ones_like(lse_r)
        else:
            m_new = torch.maximum(m_run, lse_r)
            o_num = o_num * torch.exp(m_run - m_new) \
                + o_r * torch.exp(lse_r - m_new)
            w_sum = w_sum * torch.exp(m_run - m_new) + torch.exp(lse_r - m_new)
            m_run = m_new
    return (o_num / w_sum).to(q.dtype)
