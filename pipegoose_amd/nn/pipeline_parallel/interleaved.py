"""Interleaved (virtual-stage) 1F1B pipeline schedule + engine.

Each rank holds ``v`` model chunks; global stage ``c*p + r`` lives on rank
``r`` as chunk ``c`` (Megatron-LM interleaving, arXiv:2104.04473 §2.2 — the
bubble shrinks from (p-1)/m to (p-1)/(v*m)).  The reference has no
interleaving (its GPipe engine is nn/pipeline_parallel/pipeline_engine.py);
this is a beyond-reference capability.

Transport correctness without tags: for every ordered rank pair the
messages are single-direction and both endpoints issue them in the SAME
rank-independent formula order (forwards: groups of ``p`` microbatches per
chunk), so FIFO matching per (src, dst) suffices — see the engine notes.

Status: CPU/gloo-validated (oracle-parity tests, pp2×v2); RCCL multi-GPU
validation is scheduled for round 2 — the default schedule remains "1f1b".
"""
from typing import List, Optional, Tuple

import torch

from pipegoose_amd.distributed.p2p import P2P
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.pipeline_parallel import microbatch
from pipegoose_amd.nn.pipeline_parallel._comm import PipelineP2P


def _fwd_seq(v: int, m: int, p: int) -> List[Tuple[int, int]]:
    """Rank-independent forward order: [(chunk, mb), ...].  Microbatches
    advance in blocks of p per chunk, cycling chunks (Megatron §2.2)."""
    assert m % p == 0, "interleaved 1F1B needs n_microbatches % pp == 0"
    seq = []
    for block in range(m // p):
        for c in range(v):
            for i in range(p):
                seq.append((c, block * p + i))
    return seq


def _bwd_seq(v: int, m: int, p: int) -> List[Tuple[int, int]]:
    """Backward order: same block pattern with chunks reversed."""
    seq = []
    for block in range(m // p):
        for c in reversed(range(v)):
            for i in range(p):
                seq.append((c, block * p + i))
    return seq


def interleaved_schedule(rank: int, p: int, v: int, m: int
                         ) -> List[Tuple[str, int, int]]:
    """Per-rank action list [("F"|"B", chunk, mb), ...]: warmup forwards,
    1F1B steady state, drain backwards.  Warmup per Megatron:
    (p - rank - 1) * 2 + (v - 1) * p, capped at the total forward count."""
    fwd = _fwd_seq(v, m, p)
    bwd = _bwd_seq(v, m, p)
    total = len(fwd)
    warmup = min((p - rank - 1) * 2 + (v - 1) * p, total)
    acts: List[Tuple[str, int, int]] = []
    for c, i in fwd[:warmup]:
        acts.append(("F", c, i))
    fi, bi = warmup, 0
    while fi < total:
        acts.append(("F", *fwd[fi])); fi += 1
        acts.append(("B", *bwd[bi])); bi += 1
    while bi < total:
        acts.append(("B", *bwd[bi])); bi += 1
    return acts


class InterleavedPipelineEngine:
    """1F1B over v local chunks per rank.

    Recvs are PREFETCHED on the per-direction recv streams (r2): every
    activation recv of a run has the same (shape, dtype) and per-(src,dst)
    FIFO order equals the schedule order on both endpoints, so after the
    first (codec-negotiated) recv the engine posts every remaining forward
    irecv at once, and the first grad recv posts the remaining grad
    irecvs — P2P latency overlaps compute exactly as in PipelineEngine."""

    def __init__(self, chunks: List[torch.nn.Module],
                 parallel_context: ParallelContext, n_microbatches: int,
                 loss_fn=None, moe_aux_weight: float = 0.01,
                 moe_z_weight: float = 0.1):
        self.moe_aux_weight = moe_aux_weight
        self.moe_z_weight = moe_z_weight
        self.chunks = chunks
        self.v = len(chunks)
        self.pc = parallel_context
        self.m = n_microbatches
        self.loss_fn = loss_fn
        self.p2p = PipelineP2P(parallel_context)
        self.codec = P2P(parallel_context, ParallelMode.PIPELINE)
        self.rank = parallel_context.get_local_rank(ParallelMode.PIPELINE)
        self.p = parallel_context.get_world_size(ParallelMode.PIPELINE)
        self.prev_rank = parallel_context.get_prev_global_rank(ParallelMode.PIPELINE)
        self.next_rank = parallel_context.get_next_global_rank(ParallelMode.PIPELINE)
        # negotiation is per direction: a rank learns the activation shape
        # from its first codec RECV; its first SEND goes through the codec
        # so the peer can learn it too
        self._send_negotiated = False
        self._recv_negotiated = False
        self._shape = None
        self._dtype = None
        self._fwd_q = None   # prefetched (work, buf) deques, schedule order
        self._bwd_q = None

    def _pop_moe_losses(self):
        from pipegoose_amd.nn.expert_parallel import ExpertContext
        ectx = ExpertContext.get_instance()
        aux = ectx.pop_all_aux_loss()
        zl = ectx.pop_all_z_loss()
        if not aux and not zl:
            return None
        total = None
        for a in aux:
            t = a * self.moe_aux_weight
            total = t if total is None else total + t
        for z in zl:
            t = z * self.moe_z_weight
            total = t if total is None else total + t
        return total

    # stage-role helpers -----------------------------------------------------

    def _is_first_stage(self, c: int) -> bool:
        return self.rank == 0 and c == 0

    def _is_last_stage(self, c: int) -> bool:
        return self.rank == self.p - 1 and c == self.v - 1

    # ------------------------------------------------------------------- run

    def run(self, inputs, labels=None, dp=None):
        m, v = self.m, self.v
        input_mbs = microbatch.split(inputs, m) if self.rank == 0 else [None] * m
        label_mbs = microbatch.split(labels, m) \
            if (self.rank == self.p - 1 and labels is not None) else [None] * m

        from collections import deque
        sched = interleaved_schedule(self.rank, self.p, v, m)
        self._n_fwd_recv = sum(1 for kind, c, _ in sched
                               if kind == "F" and not self._is_first_stage(c))
        self._n_bwd_recv = sum(1 for kind, c, _ in sched
                               if kind == "B" and not self._is_last_stage(c))
        self._fwd_q = deque()
        self._bwd_q = deque()
        if self._recv_negotiated:
            # later runs: shapes known — post every forward recv upfront
            for _ in range(self._n_fwd_recv):
                self._fwd_q.append(self.p2p.recv_activation_async(
                    self._shape, self._dtype, self.prev_rank, channel=0))

        saved_in = [[None] * m for _ in range(v)]
        saved_out = [[None] * m for _ in range(v)]
        # MoE router aux/z losses share graph nodes with the chunk output —
        # snapshot per (chunk, mb) at forward, ride the SAME backward call
        # (identical treatment to PipelineEngine, engine.py)
        self._saved_moe = [[None] * m for _ in range(v)]
        losses: List[torch.Tensor] = []
        outputs: List[torch.Tensor] = []
        pending = []
        if dp is not None:
            dp.sync_enabled = False

        for kind, c, mb in sched:
            if kind == "F":
                self._forward(c, mb, input_mbs, saved_in, saved_out,
                              label_mbs, losses, outputs, pending)
            else:
                self._backward(c, mb, saved_in, saved_out, label_mbs,
                               losses, pending)

        for work, _payload in pending:
            work.wait()
        if dp is not None:
            dp.sync_enabled = True
            dp.sync_now()

        if self.loss_fn is not None and labels is not None:
            if self.rank == self.p - 1 and losses:
                return torch.stack(losses).sum()
            return None
        if self.rank == self.p - 1 and outputs:
            return torch.cat(outputs, dim=0)
        return None

    # ----------------------------------------------------------------- steps

    def _recv_act(self):
        if not self._recv_negotiated:
            t = self.codec.recv(self.prev_rank)
            self._shape, self._dtype = tuple(t.shape), t.dtype
            self._recv_negotiated = True
            # prefetch every remaining forward recv of this run
            for _ in range(self._n_fwd_recv - 1):
                self._fwd_q.append(self.p2p.recv_activation_async(
                    self._shape, self._dtype, self.prev_rank, channel=0))
            return t
        work, buf = self._fwd_q.popleft()
        work.wait()
        if buf.is_cuda:
            buf.record_stream(torch.cuda.current_stream())
        return buf

    def _send_act(self, out, pending):
        if not self._send_negotiated:
            self.codec.send(out, self.next_rank)
            self._send_negotiated = True
            return
        pending.append(self.p2p.send_activation(out, self.next_rank,
                                                channel=0))

    def _forward(self, c, mb, input_mbs, saved_in, saved_out, label_mbs,
                 losses, outputs, pending):
        if self._is_first_stage(c):
            x = input_mbs[mb]
        else:
            x = self._recv_act()
            x.requires_grad_(x.is_floating_point())
        saved_in[c][mb] = x
        out = self.chunks[c](x)
        saved_out[c][mb] = out
        self._saved_moe[c][mb] = self._pop_moe_losses()
        if not self._is_last_stage(c):
            self._send_act(out, pending)
        elif self.loss_fn is None or label_mbs[mb] is None:
            outputs.append(out.detach())

    def _backward(self, c, mb, saved_in, saved_out, label_mbs, losses,
                  pending):
        out = saved_out[c][mb]
        moe = self._saved_moe[c][mb]
        self._saved_moe[c][mb] = None
        if self._is_last_stage(c):
            if self.loss_fn is not None and label_mbs[mb] is not None:
                loss = self.loss_fn(out, label_mbs[mb]) / self.m
                losses.append(loss.detach())
                if moe is not None and moe.requires_grad:
                    loss = loss + moe / self.m
                loss.backward()
        else:
            if not self._bwd_q:
                # first grad recv: post the current + every remaining one on
                # the grad recv stream (all grads share the chunk-out shape)
                for _ in range(self._n_bwd_recv):
                    self._bwd_q.append(self.p2p.recv_activation_async(
                        tuple(out.shape), out.dtype, self.next_rank,
                        tag=1, channel=1))
            work, grad = self._bwd_q.popleft()
            work.wait()
            if grad.is_cuda:
                grad.record_stream(torch.cuda.current_stream())
            if moe is not None and moe.requires_grad:
                torch.autograd.backward([out, moe / self.m],
                                        grad_tensors=[grad, None])
            else:
                torch.autograd.backward(out, grad_tensors=grad)
        x = saved_in[c][mb]
        if not self._is_first_stage(c) and x is not None and x.grad is not None:
            pending.append(self.p2p.send_activation(x.grad, self.prev_rank,
                                                    tag=1, channel=1))
        saved_in[c][mb] = saved_out[c][mb] = None
