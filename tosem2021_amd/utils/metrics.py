"""Metrics registry: counters/gauges + JSONL export + Prometheus text format.

Capability parity with the corpus' observability stack the study measured
(ray metrics_agent.py:27 + prometheus_exporter.py; apollo
cyber monitor): in-process registry, JSON-lines history for offline
analysis, and a Prometheus text exposition dump.
"""
from __future__ import annotations

import json
import os
import threading
import time
from typing import Dict, List


class Metrics:
    def __init__(self):
        self._lock = threading.Lock()
        self._counters: Dict[str, float] = {}
        self._gauges: Dict[str, float] = {}
        self._history: List[dict] = []

    def inc(self, name: str, value: float = 1.0, **labels):
        key = self._key(name, labels)
        with self._lock:
            self._counters[key] = self._counters.get(key, 0.0) + value

    def set(self, name: str, value: float, **labels):
        with self._lock:
            self._gauges[self._key(name, labels)] = float(value)

    def observe_step(self, step: int, **values):
        """Record one training/pipeline step's metrics into the history."""
        rec = {"step": step, "time": time.time(), **values}
        with self._lock:
            self._history.append(rec)

    @staticmethod
    def _key(name: str, labels: dict) -> str:
        if not labels:
            return name
        lab = ",".join(f'{k}="{v}"' for k, v in sorted(labels.items()))
        return f"{name}{{{lab}}}"

    def snapshot(self) -> dict:
        with self._lock:
            return {"counters": dict(self._counters),
                    "gauges": dict(self._gauges)}

    def save_jsonl(self, path: str) -> str:
        os.makedirs(os.path.dirname(path) or ".", exist_ok=True)
        with self._lock:
            hist = list(self._history)
        with open(path, "w") as f:
            for rec in hist:
                f.write(json.dumps(rec) + "\n")
        return path

    def prometheus_text(self) -> str:
        snap = self.snapshot()
        lines = []
        for k, v in sorted(snap["counters"].items()):
            lines.append(f"{k} {v}")
        for k, v in sorted(snap["gauges"].items()):
            lines.append(f"{k} {v}")
        return "\n".join(lines) + "\n"


_GLOBAL = Metrics()


def get_metrics() -> Metrics:
    return _GLOBAL
