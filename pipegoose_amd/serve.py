"""HTTP inference serving with dynamic micro-batching.

Beyond-reference capability (xrsrke/pipegoose has no serving path at
all): a small production-shaped serving stack over the native model
families' ``generate()`` — which on GPU auto-routes greedy tp=1 decode
through the cached hipGraph decoder (models/generation.py).

Components:

- ``BatchingEngine``: thread-safe request queue + background worker that
  coalesces compatible requests (same prompt length / decode params)
  into one batched ``generate()`` call — decode cost is launch-bound, so
  batching is where serving throughput comes from (tools/decode_bench.py:
  B8 decode is ~7x the per-sequence rate of B1).
- ``make_app``: FastAPI app with POST /generate and GET /healthz.
- ``serve``: run the whole thing.  Under tensor parallelism rank 0 owns
  the HTTP endpoint and broadcasts each batch over the TENSOR group;
  the other ranks sit in ``worker_loop`` running the same collective
  ``generate()`` calls (every TP rank must participate in the sharded
  forward).

Usage (single GPU)::

    from pipegoose_amd.serve import serve
    serve(model, host="0.0.0.0", port=8000)

TP serving (one process per GPU, e.g. torchrun --nproc-per-node 2)::

    serve(model, parallel_context=ctx)   # rank 0 serves, others loop

Requests (ids in, ids out — tokenization is the caller's business unless
a HF tokenizer is passed to make_app)::

    POST /generate {"input_ids": [[1,2,3]], "max_new_tokens": 16}
    -> {"output_ids": [[1,2,3, ...]]}
"""
import queue
import threading
import time
from dataclasses import dataclass, field
from typing import List, Optional

import torch

from pipegoose_amd.distributed import functional as F
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode

# wire header for TP fan-out: [batch, prompt_len, max_new, top_k,
# eos(-1=none), temperature*1e6, stop_flag]
_HDR_LEN = 7


@dataclass
class _Request:
    ids: torch.Tensor                 # [P] long
    max_new_tokens: int
    temperature: float
    top_k: int
    eos_token_id: Optional[int]
    done: threading.Event = field(default_factory=threading.Event)
    result: Optional[torch.Tensor] = None   # [P + new] long
    error: Optional[str] = None

    def key(self):
        return (self.ids.numel(), self.max_new_tokens, self.temperature,
                self.top_k, self.eos_token_id)


class BatchingEngine:
    """Coalesce compatible requests into batched generate() calls.

    Requests are compatible when prompt length and decode parameters
    match (the native models take unpadded [B, P] prompts).  The worker
    waits up to ``max_wait_ms`` after the first request of a batch for
    more to arrive, then runs up to ``max_batch`` of them in one call.
    """

    def __init__(self, model, parallel_context: Optional[ParallelContext]
                 = None, max_batch: int = 8, max_wait_ms: float = 5.0):
        self.model = model
        self.ctx = parallel_context or getattr(model, "parallel_context",
                                               None)
        self.device = next(model.parameters()).device
        self.max_batch = max_batch
        self.max_wait_ms = max_wait_ms
        self._q: "queue.Queue[_Request]" = queue.Queue()
        self._pending: List[_Request] = []
        self._stop = threading.Event()
        self._worker = threading.Thread(target=self._run, daemon=True)
        self._worker.start()
        self.n_batches = 0
        self.n_requests = 0

    # ------------------------------------------------------------- client API

    def submit(self, ids: torch.Tensor, max_new_tokens: int = 20,
               temperature: float = 0.0, top_k: int = 0,
               eos_token_id: Optional[int] = None,
               timeout: float = 300.0) -> torch.Tensor:
        """Blocking: returns the full [P + new] sequence for one prompt."""
        req = _Request(ids=ids.reshape(-1).to(torch.long),
                       max_new_tokens=max_new_tokens,
                       temperature=temperature, top_k=top_k,
                       eos_token_id=eos_token_id)
        self._q.put(req)
        if not req.done.wait(timeout):
            raise TimeoutError("generate timed out")
        if req.error is not None:
            raise RuntimeError(req.error)
        return req.result

    def shutdown(self):
        self._stop.set()
        self._q.put(None)           # wake the worker
        self._worker.join(timeout=10)
        if self.ctx is not None and self._tp_world() > 1:
            self._broadcast_stop()

    # ---------------------------------------------------------------- worker

    def _tp_world(self) -> int:
        return self.ctx.get_world_size(ParallelMode.TENSOR) \
            if self.ctx is not None else 1

    def _take_batch(self) -> List[_Request]:
        first = self._pending.pop(0) if self._pending else self._q.get()
        if first is None:
            return []
        batch = [first]
        deadline = time.monotonic() + self.max_wait_ms / 1e3
        while len(batch) < self.max_batch:
            left = deadline - time.monotonic()
            try:
                r = self._q.get(timeout=max(left, 0)) if left > 0 \
                    else self._q.get_nowait()
            except queue.Empty:
                break
            if r is None:
                return batch
            if r.key() == first.key():
                batch.append(r)
            else:
                self._pending.append(r)   # next batch
        return batch

    def _run(self):
        while not self._stop.is_set():
            batch = self._take_batch()
            if not batch:
                continue
            try:
                out = self._generate(batch)
                for i, r in enumerate(batch):
                    r.result = out[i].cpu()
            except Exception as e:          # surface, don't kill the loop
                for r in batch:
                    r.error = f"{type(e).__name__}: {e}"
            finally:
                for r in batch:
                    r.done.set()
                self.n_batches += 1
                self.n_requests += len(batch)

    def _generate(self, batch: List[_Request]) -> torch.Tensor:
        r0 = batch[0]
        ids = torch.stack([r.ids for r in batch]).to(self.device)
        if self._tp_world() > 1:
            self._broadcast_work(ids, r0)
        with torch.no_grad():
            return self.model.generate(
                ids, max_new_tokens=r0.max_new_tokens,
                temperature=r0.temperature, top_k=r0.top_k,
                eos_token_id=r0.eos_token_id)

    # ------------------------------------------------------- TP fan-out wire

    def _src(self) -> int:
        return self.ctx.get_ranks_in_group(ParallelMode.TENSOR)[0]

    def _broadcast_work(self, ids: torch.Tensor, r0: _Request):
        hdr = torch.tensor(
            [ids.size(0), ids.size(1), r0.max_new_tokens, r0.top_k,
             -1 if r0.eos_token_id is None else r0.eos_token_id,
             int(r0.temperature * 1e6), 0],
            dtype=torch.long, device=ids.device)
        F.broadcast(hdr, src=self._src(), parallel_context=self.ctx,
                    parallel_mode=ParallelMode.TENSOR)
        F.broadcast(ids, src=self._src(), parallel_context=self.ctx,
                    parallel_mode=ParallelMode.TENSOR)

    def _broadcast_stop(self):
        hdr = torch.zeros(_HDR_LEN, dtype=torch.long, device=self.device)
        hdr[-1] = 1
        F.broadcast(hdr, src=self._src(), parallel_context=self.ctx,
                    parallel_mode=ParallelMode.TENSOR)


def worker_loop(model, parallel_context: ParallelContext):
    """Non-rank-0 TP ranks: receive batches and join the collective
    generate() calls until a stop header arrives."""
    ctx = parallel_context
    device = next(model.parameters()).device
    src = ctx.get_ranks_in_group(ParallelMode.TENSOR)[0]
    while True:
        hdr = torch.zeros(_HDR_LEN, dtype=torch.long, device=device)
        F.broadcast(hdr, src=src, parallel_context=ctx,
                    parallel_mode=ParallelMode.TENSOR)
        b, p, max_new, top_k, eos, temp_u, stop = hdr.tolist()
        if stop:
            return
        ids = torch.zeros(b, p, dtype=torch.long, device=device)
        F.broadcast(ids, src=src, parallel_context=ctx,
                    parallel_mode=ParallelMode.TENSOR)
        with torch.no_grad():
            model.generate(ids, max_new_tokens=max_new,
                           temperature=temp_u / 1e6, top_k=top_k,
                           eos_token_id=None if eos < 0 else eos)


# ------------------------------------------------------------------ HTTP app

def make_app(engine: BatchingEngine, tokenizer=None):
    """FastAPI app over a BatchingEngine.  ``tokenizer`` (optional, HF
    interface) enables text-in/text-out via the "prompt" field."""
    from fastapi import FastAPI, HTTPException
    from pydantic import BaseModel

    class GenRequest(BaseModel):
        input_ids: Optional[List[List[int]]] = None
        prompt: Optional[str] = None
        max_new_tokens: int = 20
        temperature: float = 0.0
        top_k: int = 0
        eos_token_id: Optional[int] = None

    app = FastAPI(title="pipegoose_amd serving")
    app.state.engine = engine

    @app.get("/healthz")
    def healthz():
        return {"status": "ok", "device": str(engine.device),
                "batches": engine.n_batches, "requests": engine.n_requests}

    @app.post("/generate")
    def generate(req: GenRequest):
        if req.prompt is not None:
            if tokenizer is None:
                raise HTTPException(400, "no tokenizer configured; "
                                    "send input_ids")
            rows = [tokenizer(req.prompt)["input_ids"]]
        elif req.input_ids:
            rows = req.input_ids
        else:
            raise HTTPException(400, "input_ids or prompt required")
        outs = []
        for row in rows:
            out = engine.submit(torch.tensor(row, dtype=torch.long),
                                max_new_tokens=req.max_new_tokens,
                                temperature=req.temperature,
                                top_k=req.top_k,
                                eos_token_id=req.eos_token_id)
            outs.append(out.tolist())
        resp = {"output_ids": outs}
        if req.prompt is not None:
            resp["text"] = [tokenizer.decode(o) for o in outs]
        return resp

    return app


def serve(model, host: str = "127.0.0.1", port: int = 8000,
          parallel_context: Optional[ParallelContext] = None,
          tokenizer=None, max_batch: int = 8, max_wait_ms: float = 5.0):
    """Serve ``model``.  Rank 0 of the TENSOR group runs the HTTP
    endpoint; other ranks block in worker_loop until shutdown."""
    ctx = parallel_context or getattr(model, "parallel_context", None)
    if ctx is not None and ctx.get_world_size(ParallelMode.TENSOR) > 1 \
            and ctx.get_local_rank(ParallelMode.TENSOR) != 0:
        worker_loop(model, ctx)
        return
    engine = BatchingEngine(model, parallel_context=ctx,
                            max_batch=max_batch, max_wait_ms=max_wait_ms)
    import uvicorn
    try:
        uvicorn.run(make_app(engine, tokenizer), host=host, port=port,
                    log_level="warning")
    finally:
        engine.shutdown()
