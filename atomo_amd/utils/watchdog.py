"""Straggler / hang detection.

The reference's straggler mitigation is a master-sent kill signal that
workers poll mid-backward plus a timeout-decorated backward
(resnet_split.py:458-684, lenet.py:158-225; unwired in its main path).  In a
collective world a straggling rank stalls everyone inside RCCL, so the
equivalent machinery is a per-rank watchdog: a daemon thread that checks
steps keep completing within a deadline and, on expiry, logs a structured
record and kills the process (turning a silent collective hang into a fast,
attributable failure — torch.distributed's own timeout then releases the
other ranks).  ``action="warn"`` only logs (used by tests and for
monitoring)."""

from __future__ import annotations

import json
import os
import sys
import threading
import time
from typing import Callable, Optional


class StepWatchdog:
    def __init__(
        self,
        timeout_s: float,
        action: str = "abort",
        rank: int = 0,
        on_expire: Optional[Callable[[float], None]] = None,
        poll_s: Optional[float] = None,
    ):
        assert action in ("abort", "warn")
        self.timeout_s = float(timeout_s)
        self.action = action
        self.rank = rank
        self.on_expire = on_expire
        self.poll_s = poll_s or max(0.05, self.timeout_s / 4)
        self._last = time.monotonic()
        self._stop = threading.Event()
        self._fired = 0
        self._thread = threading.Thread(target=self._run, daemon=True)

    # heartbeat: call once per completed global step
    def step(self) -> None:
        self._last = time.monotonic()

    @property
    def fired(self) -> int:
        return self._fired

    def start(self) -> "StepWatchdog":
        self._thread.start()
        return self

    def stop(self) -> None:
        self._stop.set()

    def _run(self) -> None:
        while not self._stop.wait(self.poll_s):
            age = time.monotonic() - self._last
            if age > self.timeout_s:
                self._fired += 1
                rec = {
                    "log": "watchdog",
                    "rank": self.rank,
                    "stalled_s": round(age, 3),
                    "timeout_s": self.timeout_s,
                    "action": self.action,
                }
                print(json.dumps(rec), file=sys.stderr, flush=True)
                if self.on_expire is not None:
                    self.on_expire(age)
                if self.action == "abort":
                    sys.stderr.flush()
                    os._exit(124)
                self._last = time.monotonic()  # warn: rearm
