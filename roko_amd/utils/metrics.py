"""Observability: structured throughput counters and profiler ranges.

The reference has no tracing or metrics at all (SURVEY.md §5.1 — progress
prints and tqdm only). This module provides:

  * ``Meter`` — cheap windowed throughput counters (windows/s, bases/s —
    the BASELINE.json metrics) with a one-line JSON dump per report, so
    long polishing/training runs emit machine-readable progress;
  * ``trace_range`` — rocprof-visible named ranges (rocTX via torch's nvtx
    shim, which maps onto roctx on ROCm); no-ops when unavailable so the
    hot path never pays for disabled tracing.
"""

from __future__ import annotations

import json
import sys
import time
from contextlib import contextmanager
from typing import Dict, Optional

import torch


@contextmanager
def trace_range(name: str):
    """Named range visible in rocprofv3 marker traces (roctx)."""
    try:
        torch.cuda.nvtx.range_push(name)
        pushed = True
    except Exception:
        pushed = False
    try:
        yield
    finally:
        if pushed:
            try:
                torch.cuda.nvtx.range_pop()
            except Exception:
                pass


class Meter:
    """Windowed throughput counters with JSONL reporting.

    >>> m = Meter("inference", report_every=5.0, stream=sys.stderr)
    >>> m.add(windows=128, bases=3840)   # per batch
    ...
    >>> m.close()                        # final report
    """

    def __init__(self, stage: str, report_every: float = 10.0, stream=None,
                 rank: int = 0):
        self.stage = stage
        self.report_every = report_every
        self.stream = stream if stream is not None else sys.stderr
        self.rank = rank
        self.t0 = time.perf_counter()
        self.last_report = self.t0
        self.totals: Dict[str, float] = {}
        self.window: Dict[str, float] = {}

    def add(self, **counts: float) -> None:
        for k, v in counts.items():
            self.totals[k] = self.totals.get(k, 0.0) + v
            self.window[k] = self.window.get(k, 0.0) + v
        now = time.perf_counter()
        if now - self.last_report >= self.report_every:
            self._emit(now)

    def _emit(self, now: float) -> None:
        dt = max(now - self.last_report, 1e-9)
        rec = {
            "stage": self.stage,
            "rank": self.rank,
            "elapsed_s": round(now - self.t0, 3),
            **{f"{k}_total": self.totals[k] for k in sorted(self.totals)},
            **{f"{k}_per_s": round(self.window.get(k, 0.0) / dt, 1)
               for k in sorted(self.window)},
        }
        print(json.dumps(rec), file=self.stream, flush=True)
        self.window = {}
        self.last_report = now

    def close(self) -> Optional[dict]:
        now = time.perf_counter()
        dt = max(now - self.t0, 1e-9)
        rec = {
            "stage": self.stage,
            "rank": self.rank,
            "elapsed_s": round(now - self.t0, 3),
            **{f"{k}_total": self.totals[k] for k in sorted(self.totals)},
            **{f"{k}_per_s_avg": round(self.totals[k] / dt, 1)
               for k in sorted(self.totals)},
            "final": True,
        }
        print(json.dumps(rec), file=self.stream, flush=True)
        return rec
