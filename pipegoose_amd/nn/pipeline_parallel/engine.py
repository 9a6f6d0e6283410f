"""Pipeline engine: executes a 1F1B (or GPipe) schedule with RCCL P2P.

Redesign of the reference's engine (nn/pipeline_parallel/pipeline_engine.py):
no RPC, no worker threads, no per-clock global barriers, no autograd-side-
effect backward (the part SURVEY says not to imitate — _job/creator.py:162-277).
Each rank walks its own action list; ordering is enforced by the blocking P2P
matches themselves (forward acts flow down, grads flow up — distinct channel
directions, so RCCL needs no tags).  The first microbatch of a run uses the
typed P2P codec to negotiate shape/dtype; steady state sends raw payloads.

Backward IS an explicit engine phase: saved (input, output) per microbatch,
`torch.autograd.backward(output, grad)` then ship `input.grad` upstream.
"""
from typing import Callable, List, Optional

import torch

from pipegoose_amd.distributed.p2p import P2P
from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode
from pipegoose_amd.nn.pipeline_parallel import microbatch
from pipegoose_amd.nn.pipeline_parallel._comm import PipelineP2P
from pipegoose_amd.nn.pipeline_parallel.scheduler import (
    GPipeScheduler,
    JobType,
    OneFOneBScheduler,
)


class PipelineEngine:
    def __init__(
        self,
        stage_module: torch.nn.Module,
        parallel_context: ParallelContext,
        n_microbatches: int,
        schedule: str = "1f1b",
        loss_fn: Optional[Callable] = None,
        moe_aux_weight: float = 0.01,
        moe_z_weight: float = 0.1,
    ):
        self.moe_aux_weight = moe_aux_weight
        self.moe_z_weight = moe_z_weight
        self.stage = stage_module
        self.pc = parallel_context
        self.n_microbatches = n_microbatches
        self.schedule_kind = schedule
        self.loss_fn = loss_fn
        self.p2p = PipelineP2P(parallel_context)
        self.codec = P2P(parallel_context, ParallelMode.PIPELINE)

        self.rank = parallel_context.get_local_rank(ParallelMode.PIPELINE)
        self.n_stages = parallel_context.get_world_size(ParallelMode.PIPELINE)
        self.is_first = self.rank == 0
        self.is_last = self.rank == self.n_stages - 1
        self.prev_rank = parallel_context.get_prev_global_rank(ParallelMode.PIPELINE)
        self.next_rank = parallel_context.get_next_global_rank(ParallelMode.PIPELINE)

        self._act_meta = None  # [(shape, dtype)] per boundary element

    def _actions(self):
        if self.schedule_kind == "gpipe":
            sched = GPipeScheduler(self.n_microbatches, self.n_stages)
            actions = []
            for clock in sched.get_schedule():
                actions.extend(t for t in clock if t.partition_idx == self.rank)
            return actions
        return OneFOneBScheduler(self.n_microbatches, self.n_stages) \
            .get_rank_schedule(self.rank)

    # ------------------------------------------------------------------- run

    def run(self, inputs: torch.Tensor, labels: Optional[torch.Tensor] = None,
            dp=None):
        """One training step over the pipeline.

        Returns the mean loss on the LAST stage (None elsewhere) when
        loss_fn/labels are given; otherwise the gathered outputs on the last
        stage.
        """
        m = self.n_microbatches
        input_mbs = microbatch.split(inputs, m) if self.is_first else [None] * m
        label_mbs = microbatch.split(labels, m) \
            if (self.is_last and labels is not None) else [None] * m

        saved_in: List[Optional[torch.Tensor]] = [None] * m
        saved_out: List[Optional[torch.Tensor]] = [None] * m
        self._saved_moe = [None] * m  # per-microbatch weighted aux+z losses
        acts = self._actions()
        # recv prefetch bookkeeping: posts must follow the consumption order
        # (= this rank's schedule order = the peer's send order)
        self._fwd_order = [t.microbatch_idx for t in acts
                           if t.job_type == JobType.FORWARD]
        self._bwd_order = [t.microbatch_idx for t in acts
                           if t.job_type == JobType.BACKWARD]
        self._fwd_prefetch = {}
        self._bwd_prefetch = {}
        losses: List[torch.Tensor] = []
        outputs: List[torch.Tensor] = []
        pending = []  # (work, payload) keep-alives

        if dp is None:
            dp = getattr(self.stage, "_dp_wrapper", None)
        if dp is not None:
            # one bucketed sync at the tail instead of per-microbatch
            # re-reductions (those are numerically idempotent but cost
            # n_microbatches x the DP wire traffic)
            dp.sync_enabled = False

        from pipegoose_amd.utils.tracing import trace_range
        for task in acts:
            mb = task.microbatch_idx
            with trace_range(f"pp:{task.job_type.name.lower()}:mb{mb}"):
                self._run_task(task, mb, input_mbs, label_mbs, saved_in,
                               saved_out, losses, outputs, pending, m)

        for work, _payload in pending:
            work.wait()

        if dp is not None:
            dp.sync_enabled = True
            dp.sync_now()

        if self.loss_fn is not None and labels is not None:
            if self.is_last and losses:
                return torch.stack(losses).sum()
            return None
        if self.is_last:
            if not outputs:
                return None
            if isinstance(outputs[0], tuple):  # fx stage with tuple output
                return tuple(torch.cat(parts, dim=0)
                             for parts in zip(*outputs))
            return torch.cat(outputs, dim=0)
        return None

    def _run_task(self, task, mb, input_mbs, label_mbs, saved_in, saved_out,
                  losses, outputs, pending, m):
        has_loss = self.loss_fn is not None and label_mbs[mb] is not None
        if task.job_type == JobType.FORWARD:
            if self.is_first:
                x = input_mbs[mb]
            else:
                x = self._recv_forward(mb)
                for el in _as_tuple(x):
                    el.requires_grad_(el.is_floating_point())
            saved_in[mb] = x
            # fx-partitioned stages exchange a TUPLE of live values (skip
            # connections etc. — partitioner.FxUniformPartitioner); plain
            # stages keep the single-tensor fast path
            out = self.stage(*x) if isinstance(x, tuple) else self.stage(x)
            saved_out[mb] = out
            # Snapshot MoE router aux/z losses pushed during THIS forward.
            # They share graph nodes with `out` (the gate reads the stage's
            # hidden states), so they must ride the SAME backward call as
            # this microbatch — a separate end-of-run backward would hit an
            # already-freed graph.  Under PP the last stage's ExpertLoss
            # wrapper only ever sees its own stage's losses; the engine owns
            # the weights for every stage instead (moe_aux_weight/moe_z_weight).
            self._saved_moe[mb] = self._pop_moe_losses()
            if not self.is_last:
                pending.extend(self._send_forward(out, mb))
            elif not has_loss:
                outputs.append(tuple(o.detach() for o in out)
                               if isinstance(out, tuple) else out.detach())
        else:  # BACKWARD
            out = saved_out[mb]
            moe = self._saved_moe[mb]
            if self.is_last:
                if has_loss:
                    loss = self.loss_fn(out, label_mbs[mb]) / m
                    losses.append(loss.detach())
                    if moe is not None and moe.requires_grad:
                        loss = loss + moe / m
                    loss.backward()
                # inference-only: nothing to do
            else:
                grads = self._recv_backward(mb, out)
                # pair grads with the float elements; drop grads for
                # elements that don't require grad (e.g. detached paths)
                outs_t = _as_tuple(out)
                bwd_outs, bwd_grads = [], []
                gi = 0
                for o in outs_t:
                    if o.is_floating_point():
                        g = grads[gi]
                        gi += 1
                        if o.requires_grad:
                            bwd_outs.append(o)
                            bwd_grads.append(g)
                if moe is not None and moe.requires_grad:
                    bwd_outs.append(moe / m)
                    bwd_grads.append(None)
                torch.autograd.backward(bwd_outs, grad_tensors=bwd_grads)
            x = saved_in[mb]
            if not self.is_first and x is not None:
                gsend = [el.grad if el.grad is not None else
                         torch.zeros_like(el)
                         for el in _as_tuple(x) if el.is_floating_point()]
                if gsend:
                    pending.extend(self._send_backward(gsend, mb))
            saved_in[mb] = saved_out[mb] = None  # free activations
            self._saved_moe[mb] = None

    def _pop_moe_losses(self):
        """Drain ExpertContext into one weighted scalar (or None)."""
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

    # ------------------------------------------------------------- transport
    # A stage boundary is a tuple of tensors (singleton for plain stages).
    # Microbatch 0 negotiates count/shape/dtype via the typed codec; every
    # later microbatch moves raw payloads element by element on the
    # per-direction streams.  Backward carries one grad per FLOAT forward
    # element (zeros where autograd produced none).

    def _send_forward(self, out, mb):
        outs = _as_tuple(out)
        if mb == 0:
            self.codec.send_many(outs, self.next_rank)
            return [(_NullWork(), None)]
        return [self.p2p.send_activation(o, self.next_rank, channel=0)
                for o in outs]

    def _recv_forward(self, mb):
        if mb == self._fwd_order[0]:
            # first forward: negotiate shapes, then prefetch every remaining
            # forward recv on the recv stream IN SCHEDULE ORDER (= the
            # sender's send order) — P2P latency overlaps compute
            ts = self.codec.recv_many(self.prev_rank)
            self._act_meta = [(tuple(t.shape), t.dtype) for t in ts]
            for nxt in self._fwd_order[1:]:
                self._fwd_prefetch[nxt] = [
                    self.p2p.recv_activation_async(sh, dt, self.prev_rank,
                                                   channel=0)
                    for sh, dt in self._act_meta]
            return ts[0] if len(ts) == 1 else tuple(ts)
        bufs = []
        for work, buf in self._fwd_prefetch.pop(mb):
            work.wait()  # on CUDA: a stream dependency, not a host stall
            if buf.is_cuda:
                buf.record_stream(torch.cuda.current_stream())
            buf.requires_grad_(buf.is_floating_point())
            bufs.append(buf)
        return bufs[0] if len(bufs) == 1 else tuple(bufs)

    def _send_backward(self, grads, mb):
        return [self.p2p.send_activation(g, self.prev_rank, channel=1)
                for g in grads]

    def _grad_meta(self, out):
        return [(tuple(o.shape), o.dtype) for o in _as_tuple(out)
                if o.is_floating_point()]

    def _recv_backward(self, mb, out):
        meta = self._grad_meta(out)
        if mb in self._bwd_prefetch:
            bufs = []
            for work, buf in self._bwd_prefetch.pop(mb):
                work.wait()
                if buf.is_cuda:
                    buf.record_stream(torch.cuda.current_stream())
                bufs.append(buf)
            return bufs
        # first backward: post the CURRENT recv first (the peer sends it
        # first), then prefetch the rest in schedule order
        cur = [self.p2p.recv_activation_async(sh, dt, self.next_rank,
                                              channel=1) for sh, dt in meta]
        started = False
        for nxt in self._bwd_order:
            if nxt == mb:
                started = True
                continue
            if started:
                self._bwd_prefetch[nxt] = [
                    self.p2p.recv_activation_async(sh, dt, self.next_rank,
                                                   channel=1)
                    for sh, dt in meta]
        bufs = []
        for work, buf in cur:
            work.wait()
            if buf.is_cuda:
                buf.record_stream(torch.cuda.current_stream())
            bufs.append(buf)
        return bufs


def _as_tuple(x):
    return x if isinstance(x, tuple) else (x,)


class _NullWork:
    def wait(self):
        pass
