"""Lightweight step tracing: hipEvent-timed phase spans -> Chrome trace.

The reference has no observability beyond prints; this
gives the runner per-phase GPU timings (forward, backward, grad-sync,
optimizer) with negligible overhead, exported as a chrome://tracing JSON
and a per-phase summary.

Enable with METIS_TRACE=/path/trace.json (PlanRunner picks it up), or use
``StepTracer`` directly.
"""

from __future__ import annotations

import json
import os
import time
from contextlib import contextmanager
from typing import Dict, List, Optional

import torch


class StepTracer:
    def __init__(self, path: Optional[str] = None, rank: int = 0):
        self.path = path
        self.rank = rank
        self.enabled = path is not None
        self._events: List = []      # (name, start_ev/ts, end_ev/ts, step)
        self._step = 0
        self._use_cuda = torch.cuda.is_available()

    @contextmanager
    def span(self, name: str):
        if not self.enabled:
            yield
            return
        if self._use_cuda:
            start = torch.cuda.Event(enable_timing=True)
            end = torch.cuda.Event(enable_timing=True)
            start.record()
            yield
            end.record()
        else:
            start = time.perf_counter()
            yield
            end = time.perf_counter()
        self._events.append((name, start, end, self._step))

    def next_step(self) -> None:
        self._step += 1

    def summary(self) -> Dict[str, float]:
        """Total ms per phase name (synchronizes)."""
        if self._use_cuda:
            torch.cuda.synchronize()
        totals: Dict[str, float] = {}
        for name, start, end, _ in self._events:
            ms = (
                start.elapsed_time(end) if self._use_cuda
                else (end - start) * 1000.0
            )
            totals[name] = totals.get(name, 0.0) + ms
        return totals

    def export(self) -> None:
        """Write a chrome://tracing JSON (phase spans, us timestamps)."""
        if not self.enabled or not self._events:
            return
        if self._use_cuda:
            torch.cuda.synchronize()
        trace = []
        cursor: Dict[int, float] = {}
        for name, start, end, step in self._events:
            ms = (
                start.elapsed_time(end) if self._use_cuda
                else (end - start) * 1000.0
            )
            t0 = cursor.get(step, step * 1e6)
            trace.append({
                "name": name, "ph": "X", "pid": self.rank, "tid": step,
                "ts": t0, "dur": ms * 1000.0, "args": {"step": step},
            })
            cursor[step] = t0 + ms * 1000.0
        doc = {"traceEvents": trace,
               "metadata": {"summary_ms": self.summary()}}
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        path = self.path
        if self.rank:
            base, ext = os.path.splitext(path)
            path = f"{base}.rank{self.rank}{ext}"
        with open(path, "w") as fh:
            json.dump(doc, fh)


def tracer_from_env(rank: int) -> StepTracer:
    return StepTracer(os.environ.get("METIS_TRACE"), rank=rank)
