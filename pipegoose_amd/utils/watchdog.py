"""Training-loop hang watchdog (failure detection).

SURVEY.md §5: the reference has no failure detection at all — a desynced
collective or a stuck pipeline P2P match hangs every rank forever (RCCL
send/recv kernels spin until matched, so the Python side never raises).
This watchdog is the MI355X-native answer: the training loop calls
``tick()`` once per step; a daemon thread checks progress and, if no tick
arrives within ``timeout_s``, dumps every Python thread's stack to stderr
and (by default) hard-exits the process so the job scheduler can reclaim
the node instead of burning GPU-hours on a wedged collective.

``abort=False`` turns it into a detector only (sets ``fired``; used by
tests and by callers that want their own recovery policy).

Usage::

    wd = HangWatchdog(timeout_s=300)
    wd.start()
    for batch in loader:
        step(batch)
        wd.tick()
    wd.stop()

or as a context manager wrapping the whole loop.
"""
import faulthandler
import os
import sys
import threading
import time
from typing import Callable, Optional


class HangWatchdog:
    def __init__(self, timeout_s: float, abort: bool = True,
                 on_hang: Optional[Callable[[], None]] = None,
                 poll_s: Optional[float] = None):
        assert timeout_s > 0
        self.timeout_s = timeout_s
        self.abort = abort
        self.on_hang = on_hang
        self.poll_s = poll_s if poll_s is not None else min(1.0, timeout_s / 4)
        self.fired = False
        self._last_tick = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._steps = 0

    # ------------------------------------------------------------- lifecycle

    def start(self):
        assert self._thread is None, "watchdog already started"
        self._last_tick = time.monotonic()
        self._stop.clear()
        self._thread = threading.Thread(target=self._watch, daemon=True,
                                        name="pipegoose-watchdog")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        if self._thread is not None:
            self._thread.join(timeout=5)
            self._thread = None

    def __enter__(self):
        return self.start()

    def __exit__(self, *exc):
        self.stop()
        return False

    # ------------------------------------------------------------------ API

    def tick(self, tag: str = ""):
        """Mark forward progress (call once per training step)."""
        self._last_tick = time.monotonic()
        self._steps += 1

    # ------------------------------------------------------------- internals

    def _watch(self):
        while not self._stop.wait(self.poll_s):
            stalled = time.monotonic() - self._last_tick
            if stalled <= self.timeout_s:
                continue
            self.fired = True
            print(f"[pipegoose-watchdog] no progress for {stalled:.0f}s "
                  f"(timeout {self.timeout_s}s, {self._steps} steps seen) — "
                  f"dumping stacks", file=sys.stderr, flush=True)
            try:
                faulthandler.dump_traceback(file=sys.stderr, all_threads=True)
            except Exception:
                pass
            if self.on_hang is not None:
                try:
                    self.on_hang()
                except Exception:
                    pass
            if self.abort:
                # a hung RCCL kernel cannot be interrupted from Python;
                # exiting the process is the only way to free the GPU
                os._exit(124)
            return  # detector mode: fire once, then stand down
