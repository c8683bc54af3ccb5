"""GPU-honest timing (the reference timed kernel LAUNCHES with host clock()
and no device sync — SURVEY.md §5.1/§6; these timers do it right)."""
from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict

import torch


class PhaseTimers:
    """Named accumulating wall timers with device sync at the boundaries.

    Mirrors the reference's four per-layer accumulators
    (Sequential/Main.cpp:11) but generalized and sync-correct.
    """

    def __init__(self, sync: bool = True):
        self.totals: Dict[str, float] = defaultdict(float)
        self.counts: Dict[str, int] = defaultdict(int)
        self.sync = sync and torch.cuda.is_available()

    @contextmanager
    def phase(self, name: str):
        if self.sync:
            torch.cuda.synchronize()
        t0 = time.perf_counter()
        yield
        if self.sync:
            torch.cuda.synchronize()
        self.totals[name] += time.perf_counter() - t0
        self.counts[name] += 1

    def report(self) -> str:
        lines = []
        for k in sorted(self.totals):
            n = self.counts[k]
            tot = self.totals[k]
            lines.append(f"{k}: total {tot * 1e3:.3f} ms, n {n}, "
                         f"avg {tot / max(1, n) * 1e6:.1f} us")
        return "\n".join(lines)


class EventTimer:
    """hipEvent-based interval timing via torch.cuda.Event."""

    def __init__(self):
        self.start_ev = torch.cuda.Event(enable_timing=True)
        self.end_ev = torch.cuda.Event(enable_timing=True)

    def start(self):
        self.start_ev.record()

    def stop_ms(self) -> float:
        self.end_ev.record()
        self.end_ev.synchronize()
        return self.start_ev.elapsed_time(self.end_ev)


@contextmanager
def wall_timer():
    box = {}
    t0 = time.perf_counter()
    yield box
    box["seconds"] = time.perf_counter() - t0
