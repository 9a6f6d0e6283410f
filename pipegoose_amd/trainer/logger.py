"""Rank-aware distributed logger + throughput meter.

Reference parity: pipegoose/trainer/logger.py:4-14 was an empty stub; the
observability SURVEY.md §5 calls for lives here: a logger that only emits on
chosen ranks, and a TokensPerSecond meter the Trainer updates each step.
"""
import logging
import sys
import time
from typing import Optional

from pipegoose_amd.distributed.parallel_context import ParallelContext
from pipegoose_amd.distributed.parallel_mode import ParallelMode


class DistributedLogger:
    """stdlib-logging wrapper that prefixes the 4D rank tuple and silences
    non-selected ranks (default: global rank 0 only)."""

    def __init__(self, name: str = "pipegoose_amd",
                 parallel_context: Optional[ParallelContext] = None,
                 level: int = logging.INFO,
                 rank_zero_only: bool = True):
        self.ctx = parallel_context or ParallelContext.get_context()
        self.rank_zero_only = rank_zero_only
        self._logger = logging.getLogger(name)
        if not self._logger.handlers:
            h = logging.StreamHandler(sys.stderr)
            h.setFormatter(logging.Formatter("%(asctime)s %(levelname)s %(message)s"))
            self._logger.addHandler(h)
        self._logger.setLevel(level)
        self._logger.propagate = False

    def _prefix(self) -> str:
        if self.ctx is None:
            return ""
        parts = []
        for tag, mode in (("tp", ParallelMode.TENSOR), ("pp", ParallelMode.PIPELINE),
                          ("dp", ParallelMode.DATA)):
            if self.ctx.is_initialized(mode) and self.ctx.get_world_size(mode) > 1:
                parts.append(f"{tag}{self.ctx.get_local_rank(mode)}")
        return f"[rank{self.ctx.get_global_rank()}|{','.join(parts)}] " if parts \
            else f"[rank{self.ctx.get_global_rank()}] "

    def _should_log(self) -> bool:
        if self.ctx is None or not self.rank_zero_only:
            return True
        return self.ctx.get_global_rank() == 0

    def info(self, msg, *a):
        if self._should_log():
            self._logger.info(self._prefix() + str(msg), *a)

    def warning(self, msg, *a):
        if self._should_log():
            self._logger.warning(self._prefix() + str(msg), *a)

    def error(self, msg, *a):  # errors always emit, every rank
        self._logger.error(self._prefix() + str(msg), *a)

    def debug(self, msg, *a):
        if self._should_log():
            self._logger.debug(self._prefix() + str(msg), *a)


class ThroughputMeter:
    """Sliding tokens/sec + ms/step over the last ``window`` steps."""

    def __init__(self, window: int = 20):
        self.window = window
        self._stamps = []  # (time, tokens_cumulative)
        self.tokens = 0

    def update(self, tokens_this_step: int):
        self.tokens += tokens_this_step
        self._stamps.append((time.perf_counter(), self.tokens))
        if len(self._stamps) > self.window:
            self._stamps.pop(0)

    @property
    def tokens_per_sec(self) -> float:
        if len(self._stamps) < 2:
            return float("nan")
        (t0, n0), (t1, n1) = self._stamps[0], self._stamps[-1]
        return (n1 - n0) / max(t1 - t0, 1e-9)

    @property
    def ms_per_step(self) -> float:
        if len(self._stamps) < 2:
            return float("nan")
        (t0, _), (t1, _) = self._stamps[0], self._stamps[-1]
        return (t1 - t0) / (len(self._stamps) - 1) * 1000.0
