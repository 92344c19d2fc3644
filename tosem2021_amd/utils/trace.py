"""Chrome-timeline tracing for pipeline stages and training steps.

Capability parity with ray's profiling events (profiling.py:17
`profile(event_type)` context manager feeding the chrome timeline): emit
chrome://tracing-compatible JSON ("Trace Event Format", "X" complete events)
from nested `trace(...)` scopes, multi-process safe (pid/tid recorded).
"""
from __future__ import annotations

import json
import os
import threading
import time
from contextlib import contextmanager
from typing import Any, Dict, List


class Tracer:
    def __init__(self):
        self._events: List[Dict[str, Any]] = []
        self._lock = threading.Lock()
        self.enabled = True

    @contextmanager
    def trace(self, name: str, **args):
        if not self.enabled:
            yield
            return
        t0 = time.perf_counter_ns()
        try:
            yield
        finally:
            t1 = time.perf_counter_ns()
            ev = {
                "name": name, "ph": "X",
                "ts": t0 / 1000.0, "dur": (t1 - t0) / 1000.0,
                "pid": os.getpid(), "tid": threading.get_ident() % 2**31,
            }
            if args:
                ev["args"] = args
            with self._lock:
                self._events.append(ev)

    def instant(self, name: str, **args):
        if not self.enabled:
            return
        ev = {"name": name, "ph": "i", "ts": time.perf_counter_ns() / 1000.0,
              "pid": os.getpid(), "tid": threading.get_ident() % 2**31,
              "s": "t"}
        if args:
            ev["args"] = args
        with self._lock:
            self._events.append(ev)

    def save(self, path: str) -> str:
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        with self._lock:
            data = {"traceEvents": list(self._events)}
        with open(path, "w") as f:
            json.dump(data, f)
        return path

    def clear(self):
        with self._lock:
            self._events.clear()

    def __len__(self):
        with self._lock:
            return len(self._events)


_GLOBAL = Tracer()


def get_tracer() -> Tracer:
    return _GLOBAL


def trace(name: str, **args):
    return _GLOBAL.trace(name, **args)
