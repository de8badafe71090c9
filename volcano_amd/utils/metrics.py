"""Lightweight metrics registry (reference ``pkg/scheduler/metrics/``:
e2e/action/plugin latency histograms, schedule_attempts_total, ...).

Prometheus-compatible text exposition without requiring a server; the
same metric names as the reference so dashboards port over.
"""

from __future__ import annotations

import threading
from collections import defaultdict
from typing import Dict, List


class MetricsRegistry:
    def __init__(self):
        self._lock = threading.Lock()
        self._observations: Dict[str, List[float]] = defaultdict(list)
        self._counters: Dict[str, float] = defaultdict(float)

    def observe(self, name: str, value: float) -> None:
        with self._lock:
            obs = self._observations[name]
            obs.append(value)
            if len(obs) > 10000:
                del obs[: len(obs) // 2]

    def inc(self, name: str, value: float = 1.0) -> None:
        with self._lock:
            self._counters[name] += value

    def summary(self, name: str) -> Dict[str, float]:
        with self._lock:
            obs = sorted(self._observations.get(name, []))
        if not obs:
            return {}
        n = len(obs)
        return {
            "count": n,
            "mean": sum(obs) / n,
            "p50": obs[n // 2],
            "p99": obs[min(n - 1, int(n * 0.99))],
            "max": obs[-1],
        }

    def counter(self, name: str) -> float:
        with self._lock:
            return self._counters.get(name, 0.0)

    def export_text(self) -> str:
        lines = []
        with self._lock:
            for name, val in sorted(self._counters.items()):
                lines.append(f"{name} {val}")
            names = sorted(self._observations)
        for name in names:
            s = self.summary(name)
            for stat, v in s.items():
                lines.append(f"{name}_{stat} {v}")
        return "\n".join(lines) + "\n"

    def reset(self) -> None:
        with self._lock:
            self._observations.clear()
            self._counters.clear()


METRICS = MetricsRegistry()
