"""Per-phase timers (SURVEY §5.1 — the reference has no tracing at all).

``PhaseTimers`` accumulates wall-clock per named phase, with optional
hipEvent timing on GPU (exact device time, no sync in the hot loop until
``summary()``).  Used by the engine when ``args.enable_phase_timers``."""

from __future__ import annotations

import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List, Optional, Tuple

import torch


class PhaseTimers:
    def __init__(self, use_cuda_events: Optional[bool] = None):
        if use_cuda_events is None:
            use_cuda_events = torch.cuda.is_available()
        self.use_cuda_events = use_cuda_events
        self._cpu_acc: Dict[str, float] = defaultdict(float)
        self._counts: Dict[str, int] = defaultdict(int)
        self._pending: List[Tuple[str, torch.cuda.Event, torch.cuda.Event]] = []

    @contextmanager
    def phase(self, name: str):
        if self.use_cuda_events:
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            t0 = time.perf_counter()
            try:
                yield
            finally:
                end.record()
                self._pending.append((name, start, end))
                self._cpu_acc[name + "/host"] += time.perf_counter() - t0
                self._counts[name] += 1
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._cpu_acc[name] += time.perf_counter() - t0
                self._counts[name] += 1

    def summary(self) -> Dict[str, float]:
        """Drain pending events (synchronizes) and return ms per phase."""
        out: Dict[str, float] = {}
        if self._pending:
            torch.cuda.synchronize()
            dev_acc: Dict[str, float] = defaultdict(float)
            for name, s, e in self._pending:
                dev_acc[name] += s.elapsed_time(e)
            self._pending.clear()
            for k, v in dev_acc.items():
                out[k + "_ms"] = v
        for k, v in self._cpu_acc.items():
            out[k + "_ms"] = v * 1000.0
        out.update({k + "_count": float(v) for k, v in self._counts.items()})
        return out

    def reset(self) -> None:
        self._cpu_acc.clear()
        self._counts.clear()
        self._pending.clear()
