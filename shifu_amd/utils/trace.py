"""Lightweight step tracing: per-phase timings (fwd / loss / bwd / comm /
opt) with GPU events, aggregated per epoch and exportable as a
chrome://tracing JSON.

The reference has essentially no tracing (SURVEY §5.1 — per-epoch wall time
only); deep kernel profiling here is rocprofv3 (tools/profile_summary.py),
while this tracer answers the cheap always-on question "where does the step
go" without a profiler attached.
"""
from __future__ import annotations

import json
import os
import time
from collections import defaultdict
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch


class StepTracer:
    def __init__(self, enabled: bool = True, use_gpu_events: Optional[bool] = None,
                 keep_events: int = 2000):
        self.enabled = enabled
        self.use_gpu = (torch.cuda.is_available() if use_gpu_events is None
                        else use_gpu_events)
        self._pending: List[tuple] = []   # (phase, t0/ev0, t1/ev1, wall_ts)
        self._sums: Dict[str, float] = defaultdict(float)
        self._counts: Dict[str, int] = defaultdict(int)
        self._trace_events: List[dict] = []
        self._keep = keep_events
        self._t0 = time.time()

    @contextmanager
    def phase(self, name: str):
        if not self.enabled:
            yield
            return
        wall = time.time()
        if self.use_gpu:
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            e0.record()
            yield
            e1.record()
            self._pending.append((name, e0, e1, wall))
        else:
            t0 = time.perf_counter()
            yield
            self._pending.append((name, t0, time.perf_counter(), wall))

    def flush(self) -> None:
        """Resolve pending events (syncs once on GPU)."""
        if not self._pending:
            return
        if self.use_gpu:
            torch.cuda.synchronize()
        for name, a, b, wall in self._pending:
            ms = a.elapsed_time(b) if self.use_gpu else (b - a) * 1000.0
            self._sums[name] += ms
            self._counts[name] += 1
            if len(self._trace_events) < self._keep:
                self._trace_events.append({
                    "name": name, "ph": "X", "pid": os.getpid(), "tid": 0,
                    "ts": (wall - self._t0) * 1e6, "dur": ms * 1000.0,
                })
        self._pending.clear()

    def stats(self) -> Dict[str, Dict[str, float]]:
        self.flush()
        return {k: {"total_ms": self._sums[k], "count": self._counts[k],
                    "mean_ms": self._sums[k] / max(self._counts[k], 1)}
                for k in sorted(self._sums)}

    def summary_line(self) -> str:
        st = self.stats()
        parts = [f"{k}={v['mean_ms']:.3f}ms" for k, v in st.items()]
        return "trace: " + " ".join(parts)

    def reset(self) -> None:
        self.flush()
        self._sums.clear()
        self._counts.clear()

    def export_chrome_trace(self, path: str) -> None:
        self.flush()
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w") as f:
            json.dump({"traceEvents": self._trace_events}, f)
