"""Lightweight span tracing with chrome-trace export.

Analogue of the reference server's ProfilerService hook (SURVEY §5:
model_servers/server.cc:324,339 registers a remote-capture profiler
endpoint): here, client- and server-side stages record wall-clock spans
into a ring buffer that exports Chrome ``chrome://tracing`` /
Perfetto-compatible JSON. GPU kernel time comes from rocprofv3 (see
profiles/), host-side stage attribution comes from this.
"""
from __future__ import annotations

import json
import os
import threading
import time
from contextlib import contextmanager
from typing import List, Optional


class Tracer:
    """Thread-safe span recorder; disabled unless started (zero overhead
    when off beyond one attribute check)."""

    _global: Optional["Tracer"] = None

    def __init__(self, capacity: int = 100_000):
        self.capacity = capacity
        self._events: List[dict] = []
        self._lock = threading.Lock()
        self.enabled = False
        self._t0 = time.perf_counter()

    # -- global singleton ----------------------------------------------
    @classmethod
    def get(cls) -> "Tracer":
        if cls._global is None:
            cls._global = cls()
        return cls._global

    def start(self):
        self.enabled = True
        self._t0 = time.perf_counter()
        return self

    def stop(self):
        self.enabled = False
        return self

    @contextmanager
    def span(self, name: str, **args):
        if not self.enabled:
            yield
            return
        t0 = time.perf_counter()
        try:
            yield
        finally:
            t1 = time.perf_counter()
            ev = {
                "name": name,
                "ph": "X",
                "ts": (t0 - self._t0) * 1e6,
                "dur": (t1 - t0) * 1e6,
                "pid": os.getpid(),
                "tid": threading.get_ident() & 0xFFFF,
            }
            if args:
                ev["args"] = args
            with self._lock:
                if len(self._events) < self.capacity:
                    self._events.append(ev)

    def instant(self, name: str, **args):
        if not self.enabled:
            return
        with self._lock:
            if len(self._events) < self.capacity:
                self._events.append({
                    "name": name, "ph": "i",
                    "ts": (time.perf_counter() - self._t0) * 1e6,
                    "pid": os.getpid(),
                    "tid": threading.get_ident() & 0xFFFF,
                    "s": "t", "args": args or {}})

    def export(self, path: str):
        with self._lock:
            events = list(self._events)
        with open(path, "w") as f:
            json.dump({"traceEvents": events,
                       "displayTimeUnit": "ms"}, f)
        return len(events)

    def clear(self):
        with self._lock:
            self._events.clear()


@contextmanager
def trace_span(name: str, **args):
    with Tracer.get().span(name, **args):
        yield
