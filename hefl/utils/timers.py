"""Structured per-phase timing — the framework's tracing subsystem.

The reference instruments with bare time.time() prints (FLPyfhelin.py:224,
239,248,267,327,389; SURVEY.md section 5 tracing row). Here: a PhaseTimer
that (on GPU) brackets each phase with hipEvents so device time is measured
without host syncs inside the phase, accumulates per-phase totals, and
renders the reference-style summary. Kernel-level profiling is rocprofv3
(profiles/README.md has the recipe).
"""
from __future__ import annotations

import time
from contextlib import contextmanager
from typing import Dict, List, Optional, Tuple

import torch


class PhaseTimer:
    def __init__(self, use_gpu_events: Optional[bool] = None):
        self.use_gpu = (torch.cuda.is_available()
                        if use_gpu_events is None else use_gpu_events)
        self.wall: Dict[str, float] = {}
        self.device: Dict[str, float] = {}
        self.counts: Dict[str, int] = {}
        self._events: List[Tuple[str, object, object]] = []

    @contextmanager
    def phase(self, name: str):
        t0 = time.perf_counter()
        ev0 = ev1 = None
        if self.use_gpu:
            ev0 = torch.cuda.Event(enable_timing=True)
            ev1 = torch.cuda.Event(enable_timing=True)
            ev0.record()
        try:
            yield
        finally:
            if self.use_gpu:
                ev1.record()
                self._events.append((name, ev0, ev1))
            dt = time.perf_counter() - t0
            self.wall[name] = self.wall.get(name, 0.0) + dt
            self.counts[name] = self.counts.get(name, 0) + 1

    def collect(self):
        """Resolve pending hipEvent pairs (one sync at collection time)."""
        if self._events:
            torch.cuda.synchronize()
            for name, ev0, ev1 in self._events:
                self.device[name] = (self.device.get(name, 0.0)
                                     + ev0.elapsed_time(ev1) / 1000.0)
            self._events.clear()
        return self

    def summary(self) -> Dict[str, Dict[str, float]]:
        self.collect()
        return {k: {"wall_s": self.wall[k],
                    "device_s": self.device.get(k, float("nan")),
                    "calls": self.counts[k]} for k in self.wall}


def format_phase_table(summary: Dict[str, Dict[str, float]]) -> str:
    lines = [f"{'phase':<24} {'calls':>6} {'wall s':>10} {'device s':>10}"]
    for k, v in sorted(summary.items(), key=lambda kv: -kv[1]["wall_s"]):
        dev = v["device_s"]
        lines.append(f"{k:<24} {v['calls']:>6d} {v['wall_s']:>10.4f} "
                     f"{dev:>10.4f}" if dev == dev else
                     f"{k:<24} {v['calls']:>6d} {v['wall_s']:>10.4f} {'-':>10}")
    return "\n".join(lines)
