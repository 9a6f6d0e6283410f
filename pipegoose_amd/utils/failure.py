"""Rank-failure attribution: a heartbeat side channel over a dedicated
gloo group.

SURVEY.md §5 'Failure detection': the reference has none; round 1 added the
hang watchdog (utils/watchdog.py) which says THAT this rank is stuck but
not WHO stalled the collective.  The HeartbeatMonitor answers that: every
rank runs a daemon thread all-gathering its step counter over a gloo group
created just for this (safe next to RCCL traffic — separate communicator;
the threads' blocking all-gathers self-synchronize at the slowest rank's
pace).  When the watchdog fires, ``report()`` names the laggard ranks —
the ones whose counters stopped advancing — or reports a lost heartbeat
(process death) when the gather itself times out.

Wire-up::

    hb = HeartbeatMonitor(interval_s=2.0)
    hb.start()
    wd = HangWatchdog(timeout_s=300, on_hang=lambda: print(hb.report()))
    ...
    hb.tick()   # once per training step
"""
import datetime
import threading
import time
from typing import List, Optional

import torch
import torch.distributed as dist


class HeartbeatMonitor:
    def __init__(self, interval_s: float = 2.0, timeout_s: float = 30.0):
        assert dist.is_initialized(), "HeartbeatMonitor needs torch.distributed"
        self.world = dist.get_world_size()
        self.rank = dist.get_rank()
        self.interval_s = interval_s
        # dedicated gloo communicator: never shares state with the training
        # collectives (RCCL or gloo), so the side thread can't corrupt them
        self.group = dist.new_group(
            backend="gloo", timeout=datetime.timedelta(seconds=timeout_s))
        self._counter = 0
        self._peer_counts: List[int] = [0] * self.world
        self._last_gather: Optional[float] = None
        self._gather_error: Optional[BaseException] = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    # ---------------------------------------------------------------- control

    def start(self):
        self._thread = threading.Thread(target=self._loop, daemon=True,
                                        name="pg-heartbeat")
        self._thread.start()
        return self

    def stop(self):
        self._stop.set()
        # do NOT join: the thread may be parked inside a blocking all_gather
        # whose peers have stopped; it is a daemon and dies with the process

    def tick(self):
        """Call once per training step (cheap: one int increment)."""
        self._counter += 1

    # ------------------------------------------------------------------- loop

    def _loop(self):
        t = torch.zeros(1, dtype=torch.int64)
        bufs = [torch.zeros(1, dtype=torch.int64) for _ in range(self.world)]
        while not self._stop.is_set():
            t[0] = self._counter
            try:
                dist.all_gather(bufs, t, group=self.group)
            except BaseException as e:  # timeout/abort: peer likely dead
                self._gather_error = e
                return
            self._peer_counts = [int(b.item()) for b in bufs]
            self._last_gather = time.time()
            self._stop.wait(self.interval_s)

    # ----------------------------------------------------------------- report

    def laggards(self) -> List[int]:
        """Ranks whose counters trail the group maximum (as of the last
        completed gather)."""
        mx = max(self._peer_counts) if self._peer_counts else 0
        return [r for r, c in enumerate(self._peer_counts) if c < mx]

    def report(self) -> str:
        if self._gather_error is not None:
            return (f"[heartbeat rank {self.rank}] heartbeat LOST "
                    f"({type(self._gather_error).__name__}) — a peer process "
                    f"is likely dead; last known step counts: "
                    f"{self._peer_counts}")
        age = time.time() - self._last_gather if self._last_gather else None
        lag = self.laggards()
        if not lag:
            return (f"[heartbeat rank {self.rank}] all ranks at step "
                    f"{max(self._peer_counts, default=0)} "
                    f"(gather age {age if age is None else round(age, 1)}s) "
                    "— the stall is not step-skew (suspect a mismatched "
                    "collective inside the current step)")
        return (f"[heartbeat rank {self.rank}] step counts "
                f"{self._peer_counts}: rank(s) {lag} behind — suspect "
                "stalled/slow rank(s) "
                f"{lag} (gather age {age if age is None else round(age, 1)}s)")
