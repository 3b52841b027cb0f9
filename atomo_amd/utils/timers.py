"""Per-phase wall-clock counters.

These mirror the reference's printed per-iteration spans — worker
fetch/comp/encode/comm and Msg(MB) (distributed_worker.py:216-258), master
gather/decode (sync_replicas_master_nn.py:197-221) — but accumulate into a
dict and emit machine-parseable JSON lines, since those spans ARE the
benchmark metric."""

from __future__ import annotations

import json
import time
from collections import defaultdict
from contextlib import contextmanager

import torch


class PhaseTimers:
    def __init__(self, sync_cuda: bool = False):
        self.sync_cuda = sync_cuda
        self.totals = defaultdict(float)
        self.counts = defaultdict(int)
        self.scalars = defaultdict(float)

    def _now(self):
        if self.sync_cuda and torch.cuda.is_available():
            torch.cuda.synchronize()
        return time.perf_counter()

    @contextmanager
    def phase(self, name: str):
        t0 = self._now()
        try:
            yield
        finally:
            self.totals[name] += self._now() - t0
            self.counts[name] += 1

    def add_scalar(self, name: str, value: float):
        self.scalars[name] += value
        self.counts[name] += 1

    def reset(self):
        self.totals.clear()
        self.counts.clear()
        self.scalars.clear()

    def summary(self) -> dict:
        out = {}
        for k, v in self.totals.items():
            n = max(1, self.counts[k])
            out[f"{k}_s"] = v
            out[f"{k}_ms_avg"] = 1e3 * v / n
        for k, v in self.scalars.items():
            out[k] = v
        return out

    def emit(self, step: int, prefix: str = "atomo", **extra):
        rec = {"log": prefix, "step": step}
        rec.update(self.summary())
        rec.update(extra)
        print(json.dumps(rec), flush=True)
