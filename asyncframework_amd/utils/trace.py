"""Chrome-trace (Perfetto) event logging — the observability substrate.

The reference's substrate writes scheduler event-log JSON consumed by the
Spark History Server (reference core/.../scheduler/EventLoggingListener.scala:55,
LiveListenerBus.scala:44); the stdout timing tables are the framework-level
channel (SparkASGDThread.scala:115-119,331-338 — kept byte-exact in
utils/logfmt). This module is the MI355X rebuild of the substrate channel:
engines emit dispatch/compute/accept/reject/update events into the
chrome://tracing "Trace Event Format" (one JSON object, ``traceEvents``
array), viewable in Perfetto — alongside rocprofv3 for the kernel level.

Off the hot path: when disabled every hook is a single attribute check; when
enabled, events append to an in-memory list under a lock and are written once
at ``stop()``.

Activate with ``ASYNCAMD_TRACE=/path/trace.json`` (written at engine
shutdown) or programmatically via ``start_trace``/``stop_trace``.
"""

from __future__ import annotations

import json
import os
import threading
import time
from typing import Any, Dict, List, Optional

_T0_NS = time.perf_counter_ns()


def _now_us() -> float:
    return (time.perf_counter_ns() - _T0_NS) / 1000.0


class Tracer:
    """Collects Trace Event Format events (ph 'X' complete / 'i' instant).

    Bounded: beyond ``max_events`` (default 2M, ~hundreds of MB of JSON)
    new events are counted but dropped — a process that traces but never
    flushes (e.g. a worker rank in a week-long run) must not leak without
    bound. The saved document records how many were dropped."""

    def __init__(self, path: str = "", max_events: int = 2_000_000):
        self.path = path
        self.enabled = True
        self.max_events = max_events
        self.dropped = 0
        self._events: List[Dict[str, Any]] = []
        self._lock = threading.Lock()

    def _append(self, ev: Dict[str, Any]) -> None:
        with self._lock:
            if len(self._events) >= self.max_events:
                self.dropped += 1
                return
            self._events.append(ev)

    # pid = engine role, tid = worker id / server lane
    def complete(self, name: str, tid: int, t0_us: float, dur_us: float,
                 args: Optional[Dict[str, Any]] = None,
                 pid: str = "workers") -> None:
        ev = {"name": name, "ph": "X", "ts": t0_us, "dur": dur_us,
              "pid": pid, "tid": tid}
        if args:
            ev["args"] = args
        self._append(ev)

    def instant(self, name: str, tid: int,
                args: Optional[Dict[str, Any]] = None,
                pid: str = "server",
                ts_us: Optional[float] = None) -> None:
        ev = {"name": name, "ph": "i",
              "ts": _now_us() if ts_us is None else ts_us, "s": "t",
              "pid": pid, "tid": tid}
        if args:
            ev["args"] = args
        self._append(ev)

    def monotonic_s_to_us(self, t_s: float) -> float:
        """Convert a CLOCK_MONOTONIC-epoch stamp in seconds (C++
        steady_clock — same clock as perf_counter) to this trace's us."""
        return t_s * 1e6 - _T0_NS / 1000.0

    def now_us(self) -> float:
        return _now_us()

    def save(self, path: str = "") -> str:
        path = path or self.path
        with self._lock:
            doc = {"traceEvents": list(self._events),
                   "displayTimeUnit": "ms"}
            if self.dropped:
                doc["droppedEvents"] = self.dropped
        with open(path, "w") as f:
            json.dump(doc, f)
        return path


_tracer: Optional[Tracer] = None
_tracer_lock = threading.Lock()


def get_tracer() -> Optional[Tracer]:
    """The active tracer, or None (the common, zero-cost case). Lazily
    honors ASYNCAMD_TRACE on first call."""
    global _tracer
    if _tracer is None:
        path = os.environ.get("ASYNCAMD_TRACE", "")
        if path:
            with _tracer_lock:
                if _tracer is None:
                    _tracer = Tracer(path)
    return _tracer


def start_trace(path: str) -> Tracer:
    global _tracer
    with _tracer_lock:
        _tracer = Tracer(path)
    return _tracer


def stop_trace() -> Optional[str]:
    """Write and deactivate the current trace; returns the file path."""
    global _tracer
    with _tracer_lock:
        t, _tracer = _tracer, None
    if t is None:
        return None
    return t.save() if t.path else None
