"""Minimal real Trainer (reference Trainer.fit/train were empty stubs,
pipegoose/trainer/trainer.py:29-35).

Drives the standard loop over an iterable of batches: forward → loss →
backward → (optional grad clip) → optimizer step → scheduler step, firing
Callback events and keeping TrainerState + throughput metrics.  Works with
any combination of the parallel wrappers: the model's ``forward`` is whatever
``parallelize()`` left in place (incl. the pipeline engine), and the
optimizer may be a ZeRO-1 ``DistributedOptimizer``.
"""
import math
from typing import Callable, Iterable, Optional

import torch

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.trainer.callback import CallbackList
from pipegoose_amd.trainer.logger import DistributedLogger, ThroughputMeter
from pipegoose_amd.trainer.state import TrainerStage, TrainerState


class Trainer:
    def __init__(
        self,
        model,
        optimizer,
        loss_fn: Optional[Callable] = None,
        lr_scheduler=None,
        parallel_context: Optional[ParallelContext] = None,
        callbacks: Optional[list] = None,
        max_grad_norm: Optional[float] = None,
        log_interval: int = 10,
        logger: Optional[DistributedLogger] = None,
        hang_timeout_s: Optional[float] = None,
    ):
        self.model = model
        self.optimizer = optimizer
        self.loss_fn = loss_fn
        self.lr_scheduler = lr_scheduler
        self.ctx = parallel_context or ParallelContext.get_context()
        self.callbacks = CallbackList(callbacks)
        self.max_grad_norm = max_grad_norm
        self.log_interval = log_interval
        self.state = TrainerState()
        self.meter = ThroughputMeter()
        self.logger = logger or DistributedLogger(parallel_context=self.ctx)
        # failure detection: a desynced collective hangs every
        # rank silently; with hang_timeout_s set, fit() arms a watchdog that
        # dumps stacks and exits the process if a step stops making progress
        self.hang_timeout_s = hang_timeout_s
        self._watchdog = None

    # ----------------------------------------------------------------- steps

    def training_step(self, batch) -> torch.Tensor:
        """One forward+loss.  ``batch`` is a dict of model kwargs; if
        ``loss_fn`` is None the model must return the loss itself (e.g. a
        CausalLM called with labels=)."""
        if self.loss_fn is None:
            out = self.model(**batch)
            if out is None:
                return None  # pipeline non-last stage (engine ran backward)
            return out if torch.is_tensor(out) else out["loss"]
        labels = batch.pop("labels")
        logits = self.model(**batch)
        return self.loss_fn(logits, labels)

    def _count_tokens(self, batch) -> int:
        ids = batch.get("input_ids")
        return ids.numel() if torch.is_tensor(ids) else 0

    def train(self, batch) -> float:
        """One optimizer step on one batch; returns the loss value."""
        from pipegoose_amd.utils.tracing import trace_range
        self.callbacks.fire("on_step_start", self)
        self.model.train()
        self.optimizer.zero_grad()
        tokens = self._count_tokens(batch)
        with trace_range("trainer:forward"):
            loss = self.training_step(dict(batch))
        if loss is not None and loss.requires_grad:
            with trace_range("trainer:backward"):
                loss.backward()
        # else: a pipeline engine already ran backward internally (its
        # returned loss is detached; non-last stages return None)
        if self.max_grad_norm is not None:
            if hasattr(self.optimizer, "clip_grad_norm_"):
                # ZeRO shard-aware clip (valid grads may live only on owners)
                self.optimizer.clip_grad_norm_(self.max_grad_norm)
            else:
                torch.nn.utils.clip_grad_norm_(
                    [p for p in self.model.parameters() if p.requires_grad],
                    self.max_grad_norm)
        with trace_range("trainer:optimizer"):
            self.optimizer.step()
        if self.lr_scheduler is not None:
            self.lr_scheduler.step()
        val = float(loss.detach().float().item()) if loss is not None \
            else float("nan")
        self.state.global_step += 1
        self.state.tokens_seen += tokens
        self.state.last_loss = val
        self.meter.update(tokens)
        self.callbacks.fire("on_step_end", self, val)
        if self.log_interval and not math.isnan(val) \
                and self.state.global_step % self.log_interval == 0:
            tps = self.meter.tokens_per_sec
            self.logger.info(
                f"step {self.state.global_step} loss {val:.4f}"
                + (f" tokens/s {tps:,.0f}" if not math.isnan(tps) else ""))
        return val

    def resume_from(self, ckpt_dir: str) -> dict:
        """Restore model weights + optimizer/scheduler/step state written by
        ``CheckpointCallback`` (or save_pretrained/save_training_state) and
        continue counting from the saved step."""
        from pipegoose_amd.nn.utils import from_pretrained, load_training_state
        from_pretrained(self.model, ckpt_dir, parallel_context=self.ctx)
        payload = load_training_state(self.optimizer, ckpt_dir,
                                      parallel_context=self.ctx,
                                      lr_scheduler=self.lr_scheduler)
        self.state.global_step = payload["step"]
        return payload

    # ------------------------------------------------------------------- fit

    def fit(self, train_loader: Iterable, epochs: int = 1,
            max_steps: Optional[int] = None):
        self.state.stage = TrainerStage.TRAINING
        self.callbacks.fire("on_fit_start", self)
        if self.hang_timeout_s is not None:
            from pipegoose_amd.utils.watchdog import HangWatchdog
            import torch.distributed as dist
            self._heartbeat = None
            if dist.is_initialized() and dist.get_world_size() > 1:
                # name the stalled rank before aborting (utils/failure.py)
                from pipegoose_amd.utils.failure import HeartbeatMonitor
                self._heartbeat = HeartbeatMonitor().start()
                hb = self._heartbeat
                self._watchdog = HangWatchdog(
                    self.hang_timeout_s,
                    on_hang=lambda: print(hb.report(), flush=True)).start()
            else:
                self._watchdog = HangWatchdog(self.hang_timeout_s).start()
        try:
            for epoch in range(epochs):
                self.state.epoch = epoch
                self.callbacks.fire("on_epoch_start", self)
                for batch in train_loader:
                    self.train(batch)
                    if self._watchdog is not None:
                        self._watchdog.tick()
                        if self._heartbeat is not None:
                            self._heartbeat.tick()
                    if max_steps is not None and self.state.global_step >= max_steps:
                        break
                self.callbacks.fire("on_epoch_end", self)
                if max_steps is not None and self.state.global_step >= max_steps:
                    break
        finally:
            if self._watchdog is not None:
                self._watchdog.stop()
                self._watchdog = None
            if getattr(self, "_heartbeat", None) is not None:
                self._heartbeat.stop()
                self._heartbeat = None
        self.state.stage = TrainerStage.FINISHED
        self.callbacks.fire("on_fit_end", self)
        return self.state

    @torch.no_grad()
    def evaluate(self, eval_loader: Iterable) -> dict:
        self.callbacks.fire("on_eval_start", self)
        prev_stage = self.state.stage
        self.state.stage = TrainerStage.EVALUATING
        self.model.eval()
        total, n = 0.0, 0
        for batch in eval_loader:
            loss = self.training_step(dict(batch))
            total += float(loss.detach().float().item())
            n += 1
        metrics = {"eval_loss": total / max(n, 1)}
        self.state.metrics.update(metrics)
        self.state.stage = prev_stage
        self.callbacks.fire("on_eval_end", self, metrics)
        return metrics
