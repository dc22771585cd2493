"""Lightweight span tracing -> chrome://tracing JSON.

SURVEY.md §5 "Tracing / profiling": the reference has none; the rebuild
gets rocprofv3 for kernels (profiles/) and this host-side tracer for
scheduler/step spans. Enable with RB_TRACE=/path/out.json (the engine
wraps its step phases) or use Tracer directly; open the dump in
chrome://tracing / https://ui.perfetto.dev.
"""
from __future__ import annotations

import json
import os
import threading
import time
from contextlib import contextmanager


class Tracer:
    def __init__(self):
        self.events: list[dict] = []
        self._lock = threading.Lock()
        self.t0 = time.perf_counter()

    @contextmanager
    def span(self, name: str, **args):
        start = time.perf_counter()
        try:
            yield
        finally:
            end = time.perf_counter()
            with self._lock:
                self.events.append({
                    "name": name, "ph": "X", "cat": "engine",
                    "pid": os.getpid(),
                    "tid": threading.get_ident() % 1_000_000,
                    "ts": (start - self.t0) * 1e6,
                    "dur": (end - start) * 1e6,
                    "args": args or {}})

    def instant(self, name: str, **args):
        with self._lock:
            self.events.append({
                "name": name, "ph": "i", "s": "t", "cat": "engine",
                "pid": os.getpid(),
                "tid": threading.get_ident() % 1_000_000,
                "ts": (time.perf_counter() - self.t0) * 1e6,
                "args": args or {}})

    def dump(self, path: str) -> None:
        with self._lock:
            payload = {"traceEvents": list(self.events)}
        with open(path, "w") as f:
            json.dump(payload, f)


_GLOBAL: Tracer | None = None


def get_tracer() -> Tracer | None:
    """Process-global tracer, created when RB_TRACE is set."""
    global _GLOBAL
    if _GLOBAL is None and os.environ.get("RB_TRACE"):
        _GLOBAL = Tracer()
    return _GLOBAL


def dump_global() -> str | None:
    path = os.environ.get("RB_TRACE")
    if path and _GLOBAL is not None:
        _GLOBAL.dump(path)
        return path
    return None
