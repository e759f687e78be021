"""Native round-phase tracing (the reference has no tracing
subsystem beyond timestamped prints; the MI355X rebuild adds a per-round
phase timeline so pipeline bubbles are visible without a profiler attach).

Each Ape-X round is split into named spans (actor, gather, learn,
broadcast, ...).  On GPU every span is bracketed by ``torch.cuda.Event``
pairs on the current stream -- zero host synchronization while tracing; the
event deltas are resolved once at :meth:`dump`.  On CPU the same API falls
back to ``time.perf_counter``.  ``dump`` writes the Chrome trace-event JSON
format, so the timeline opens directly in ``chrome://tracing`` / Perfetto
alongside a rocprofv3 capture.

Usage (wired into ``train_apex --trace-file``):

    tracer = RoundTracer("trace.json")
    with tracer.span("learn"):
        ...
    tracer.dump()
"""

from __future__ import annotations

import json
import os
import time
from contextlib import contextmanager
from typing import List, Optional, Tuple

import torch


class RoundTracer:
    """Records named spans and writes a Chrome trace-event timeline."""

    def __init__(self, path: str, use_cuda: Optional[bool] = None,
                 max_spans: int = 20000, pid: int = 0):
        self.path = path
        self.pid = pid
        self.use_cuda = (torch.cuda.is_available()
                         if use_cuda is None else use_cuda)
        self.max_spans = max_spans
        # CUDA spans: (name, start_event, end_event); CPU: (name, t0, t1)
        self._cuda_spans: List[Tuple[str, torch.cuda.Event, torch.cuda.Event]] = []
        self._cpu_spans: List[Tuple[str, float, float]] = []
        self._origin: Optional[torch.cuda.Event] = None
        self._t0 = time.perf_counter()

    @property
    def full(self) -> bool:
        return len(self._cuda_spans) + len(self._cpu_spans) >= self.max_spans

    @contextmanager
    def span(self, name: str):
        if self.full:
            yield
            return
        if self.use_cuda:
            if self._origin is None:
                self._origin = torch.cuda.Event(enable_timing=True)
                self._origin.record()
            e0 = torch.cuda.Event(enable_timing=True)
            e1 = torch.cuda.Event(enable_timing=True)
            e0.record()
            try:
                yield
            finally:
                e1.record()
                self._cuda_spans.append((name, e0, e1))
        else:
            t0 = time.perf_counter()
            try:
                yield
            finally:
                self._cpu_spans.append((name, t0, time.perf_counter()))

    def events(self) -> List[dict]:
        """Resolve recorded spans to trace-event dicts (synchronizes once)."""
        out = []
        if self._cuda_spans:
            torch.cuda.synchronize()
            for name, e0, e1 in self._cuda_spans:
                ts = self._origin.elapsed_time(e0) * 1000.0  # ms -> us
                dur = e0.elapsed_time(e1) * 1000.0
                out.append({"name": name, "ph": "X", "ts": ts, "dur": dur,
                            "pid": self.pid, "tid": 0})
        for name, t0, t1 in self._cpu_spans:
            out.append({"name": name, "ph": "X",
                        "ts": (t0 - self._t0) * 1e6,
                        "dur": (t1 - t0) * 1e6,
                        "pid": self.pid, "tid": 0})
        return out

    def dump(self) -> str:
        os.makedirs(os.path.dirname(self.path) or ".", exist_ok=True)
        with open(self.path, "w") as f:
            json.dump({"traceEvents": self.events(),
                       "displayTimeUnit": "ms"}, f)
        return self.path
