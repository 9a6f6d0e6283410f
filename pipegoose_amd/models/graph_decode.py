"""hipGraph-captured greedy decode loop.

Decode at batch 8 is launch-bound (~8 ms/step on BLOOM-560M, ROADMAP.md §4):
hundreds of tiny kernel launches per step dwarf the actual compute.  A
hipGraph replays the whole step as ONE launch.  The requirements, and how
they are met here:

- constant shapes/addresses: ``GraphKVCache`` (models/kv_cache.py) — attention
  always runs over the full preallocated cache length, unfilled slots masked;
- no host values in the step: the write position is a device tensor
  (``pos_t``) advanced in-graph (`add_`); the ALiBi bias is computed from it
  (`BloomAttention._alibi_bias_graph`); greedy sampling is an in-graph argmax
  written back into the input buffer — the loop is fully self-advancing;
- every HIP kernel launches on the capture stream: all csrc bindings use
  ``at::cuda::getCurrentCUDAStream()``.

Beyond-reference capability (the reference had no decode path at all).
Falls back to eager static-cache stepping when capture is unavailable
(CPU, or ``use_graph=False``) — numerics are identical by construction and
tested (tests/nn/test_bloom_model.py, tests/ops/test_kernels_gpu.py).
"""
from typing import Optional

import torch

from pipegoose_amd.distributed.parallel_mode import ParallelMode


class GraphDecoder:
    """Greedy decoder for a (tp=1) causal LM with ``new_graph_kv_cache``.

    Usage::

        dec = GraphDecoder(model, batch_size=8, max_len=512)
        out = dec.generate(prompt_ids, max_new_tokens=128)   # [B, 128]
    """

    def __init__(self, model, batch_size: int, max_len: int,
                 use_graph: Optional[bool] = None):
        ctx = getattr(model, "parallel_context", None)
        if ctx is not None and ctx.get_world_size(ParallelMode.TENSOR) > 1:
            raise NotImplementedError(
                "GraphDecoder is single-GPU (tp=1): RCCL collectives inside "
                "graph capture are a round-2 item; use model.generate() for "
                "TP serving")
        self.model = model
        self.max_len = max_len
        p = next(model.parameters())
        self.device, self.dtype = p.device, p.dtype
        self.cache = model.new_graph_kv_cache(batch_size, max_len)
        self.ids = torch.zeros(batch_size, 1, dtype=torch.long,
                               device=self.device)
        self.use_graph = (p.device.type == "cuda") if use_graph is None \
            else use_graph
        self._graph = None

    # ------------------------------------------------------------------ step

    @torch.no_grad()
    def _step(self):
        """One self-advancing decode step (the captured region)."""
        logits, _ = self.model(self.ids, past=self.cache, use_cache=True)
        nxt = logits[:, -1].argmax(-1, keepdim=True)
        self.ids.copy_(nxt)
        self.cache.advance()

    @torch.no_grad()
    def _capture(self):
        """Run ONE real warmup step (allocator priming), then record the
        graph.  Always returns the warmup step's token — even when capture
        fails, that step advanced the state and its token must be kept."""
        try:
            s = torch.cuda.Stream()
            s.wait_stream(torch.cuda.current_stream())
        except Exception:       # no device: nothing ran, nothing to return
            self.use_graph = False
            self._graph = None
            return None
        with torch.cuda.stream(s):
            self._step()
        torch.cuda.current_stream().wait_stream(s)
        warmup_token = self.ids.clone()
        try:
            g = torch.cuda.CUDAGraph()
            with torch.cuda.graph(g):   # records, does not execute
                self._step()
            self._graph = g
        except Exception:
            # capture unsupported here: permanent eager fallback
            self.use_graph = False
            self._graph = None
        return warmup_token

    # -------------------------------------------------------------- generate

    @torch.no_grad()
    def generate(self, prompt_ids: torch.Tensor, max_new_tokens: int
                 ) -> torch.Tensor:
        """Greedy-decode ``max_new_tokens`` tokens after ``prompt_ids``
        [B, P]; returns [B, max_new_tokens]."""
        B, P = prompt_ids.shape
        assert B == self.ids.size(0), "batch size fixed at construction"
        assert P + max_new_tokens <= self.max_len, "raise max_len"
        self.model.eval()
        self.cache.reset()

        # eager prefill fills cache slots 0..P-1
        logits, _ = self.model(prompt_ids.to(self.device), past=self.cache,
                               use_cache=True)
        self.ids.copy_(logits[:, -1].argmax(-1, keepdim=True))
        self.cache.enter_graph_mode()

        tokens = [self.ids.clone()]
        remaining = max_new_tokens - 1
        if self.use_graph and remaining > 1 and self._graph is None:
            warmup_token = self._capture()
            if warmup_token is not None:  # the warmup step really ran
                tokens.append(warmup_token)
                remaining -= 1
        # a previously captured graph replays against the CURRENT buffer
        # contents — no re-capture needed across calls
        for _ in range(remaining):
            if self._graph is not None:
                self._graph.replay()
            else:
                self._step()
            tokens.append(self.ids.clone())
        return torch.cat(tokens, dim=1)
