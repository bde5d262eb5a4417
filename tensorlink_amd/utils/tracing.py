"""Step tracing: Chrome-trace (chrome://tracing / Perfetto) span export.

The reference has NO tracing subsystem — its closest artifacts are
per-token ``time.time()`` deltas (``ml/formatter.py:353``) and ad-hoc GPU
memory prints (``ml/worker.py:801-817``); SURVEY.md §5 makes a proper
profile path a deliverable of this build. Kernel-level timing comes from
``rocprofv3`` (see profiles/README.md); this module covers the layer the
kernel profiler cannot see — engine phases (prefill, decode steps,
collectives, scheduler iterations) across ranks — and emits standard
Chrome trace JSON so both land in the same timeline viewer.

Usage::

    tracer = Tracer()
    with tracer.span("prefill", batch=B):
        ...
    tracer.export("trace.json")         # open in ui.perfetto.dev

On GPU, each span also records HIP events and attaches the measured
``gpu_ms`` (device time between span entry/exit on the current stream) to
the span args at export time.
"""

from __future__ import annotations

import json
import os
import time
from contextlib import contextmanager
from typing import Any, Dict, List, Optional

import torch


class Tracer:
    def __init__(self, rank: int = 0, use_gpu_events: Optional[bool] = None):
        self.rank = rank
        self.events: List[Dict[str, Any]] = []
        self._gpu = (torch.cuda.is_available() if use_gpu_events is None
                     else use_gpu_events)
        self._pairs: List[tuple] = []       # (event_idx, ev0, ev1)
        self._t0 = time.perf_counter()

    @contextmanager
    def span(self, name: str, **args):
        ev0 = ev1 = None
        if self._gpu:
            ev0 = torch.cuda.Event(enable_timing=True)
            ev0.record()
        t0 = time.perf_counter()
        try:
            yield
        finally:
            t1 = time.perf_counter()
            if self._gpu:
                ev1 = torch.cuda.Event(enable_timing=True)
                ev1.record()
            self.events.append({
                "name": name, "ph": "X", "pid": self.rank, "tid": 0,
                "ts": (t0 - self._t0) * 1e6, "dur": (t1 - t0) * 1e6,
                "args": dict(args)})
            if ev0 is not None:
                self._pairs.append((len(self.events) - 1, ev0, ev1))

    def instant(self, name: str, **args):
        """Zero-duration marker (e.g. 'token_emitted')."""
        self.events.append({
            "name": name, "ph": "i", "s": "t", "pid": self.rank, "tid": 0,
            "ts": (time.perf_counter() - self._t0) * 1e6,
            "args": dict(args)})

    def _resolve_gpu_times(self):
        if not self._pairs:
            return
        torch.cuda.synchronize()
        for idx, ev0, ev1 in self._pairs:
            self.events[idx]["args"]["gpu_ms"] = ev0.elapsed_time(ev1)
        self._pairs.clear()

    def export(self, path: str) -> str:
        """Write Chrome trace JSON (merge-friendly: rank is the pid, so
        per-rank files can be concatenated by a viewer)."""
        self._resolve_gpu_times()
        d = os.path.dirname(path)
        if d:
            os.makedirs(d, exist_ok=True)
        with open(path, "w") as f:
            json.dump({"traceEvents": self.events,
                       "displayTimeUnit": "ms"}, f)
        return path

    def summary(self) -> Dict[str, Dict[str, float]]:
        """Per-span-name aggregate: count, total/mean wall ms."""
        self._resolve_gpu_times()
        out: Dict[str, Dict[str, float]] = {}
        for e in self.events:
            if e["ph"] != "X":
                continue
            s = out.setdefault(e["name"], {"count": 0, "total_ms": 0.0})
            s["count"] += 1
            s["total_ms"] += e["dur"] / 1e3
        for s in out.values():
            s["mean_ms"] = s["total_ms"] / s["count"]
        return out


def tracer_from_env(rank: int = 0) -> Optional[Tracer]:
    """TL_TRACE=<path-prefix> enables tracing; the export path becomes
    ``<prefix>_rank<r>.json``."""
    if not os.environ.get("TL_TRACE"):
        return None
    return Tracer(rank=rank)


def export_from_env(tracer: Optional[Tracer]) -> Optional[str]:
    if tracer is None:
        return None
    prefix = os.environ.get("TL_TRACE", "trace")
    return tracer.export(f"{prefix}_rank{tracer.rank}.json")
