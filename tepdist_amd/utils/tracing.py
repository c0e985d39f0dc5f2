"""Tracing / profiling spans.

The reference exposes: TraceMe spans on the client,
DEBUG=true per-task wall-clock logging in ExecuteTaskList, the
`[ExecutePlan Duration]` per-plan line, and Graphviz dumps of the task DAG.
Here the same surface is:
  - `trace_span(name)` context manager collecting (name, ts, dur, pid, tid)
    records, enabled by TEPDIST_TRACE=<path>; `Tracer.save()` writes a
    chrome://tracing / perfetto-compatible JSON file;
  - rocTX-style GPU ranges via torch.cuda.nvtx (roctx on ROCm) when a GPU
    is active, so rocprofv3 --marker-trace groups kernels by span;
  - per-task debug timing stays in the executors ([ExecutePlan Duration]).
"""

from __future__ import annotations

import contextlib
import json
import os
import threading
import time
from typing import List, Optional

import torch


class Tracer:
    def __init__(self, path: Optional[str] = None):
        self.path = path or os.environ.get("TEPDIST_TRACE", "")
        self.enabled = bool(self.path)
        self.events: List[dict] = []
        self._lock = threading.Lock()
        self._t0 = time.perf_counter()

    def add(self, name: str, ts: float, dur: float, pid: int = 0,
            tid: int = 0, args: dict = None):
        if not self.enabled:
            return
        with self._lock:
            self.events.append({
                "name": name, "ph": "X", "ts": ts * 1e6, "dur": dur * 1e6,
                "pid": pid, "tid": tid, "args": args or {}})

    @contextlib.contextmanager
    def span(self, name: str, pid: int = 0, tid: int = 0, args: dict = None):
        use_nvtx = torch.cuda.is_available() and torch.cuda.is_initialized()
        if use_nvtx:
            torch.cuda.nvtx.range_push(name)   # roctx range on ROCm
        t0 = time.perf_counter() - self._t0
        try:
            yield
        finally:
            t1 = time.perf_counter() - self._t0
            if use_nvtx:
                torch.cuda.nvtx.range_pop()
            self.add(name, t0, t1 - t0, pid, tid, args)

    def save(self, path: Optional[str] = None):
        p = path or self.path
        if not p or not self.events:
            return
        with self._lock:
            with open(p, "w") as f:
                json.dump({"traceEvents": self.events,
                           "displayTimeUnit": "ms"}, f)


_tracer: Optional[Tracer] = None


def get_tracer() -> Tracer:
    global _tracer
    if _tracer is None:
        _tracer = Tracer()
    return _tracer


def trace_span(name: str, **kw):
    return get_tracer().span(name, **kw)
