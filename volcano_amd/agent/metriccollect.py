"""Metric collection framework + resource-usage getter.

Reference: ``pkg/metriccollect`` (pluggable local collectors behind a
registry — framework/registry.go:31, cadvisor-style cpu/memory under
``local/``) and ``pkg/resourceusage`` (resource_usage_getter.go:35 —
the Getter the oversubscription policy consumes).

Local collectors read real host counters through psutil when available
(this image ships it); tests inject samples directly."""

from __future__ import annotations

import time
from collections import deque
from typing import Callable, Deque, Dict, List, Optional, Tuple


class MetricCollector:
    """Plugin interface (framework/interface.go analog)."""

    name = "base"

    def collect(self) -> Dict[str, float]:  # pragma: no cover - interface
        raise NotImplementedError


_REGISTRY: Dict[str, Callable[[], MetricCollector]] = {}


def register_collector(name: str):
    """framework/registry.go:31 RegisterMetricCollect."""
    def deco(factory):
        _REGISTRY[name] = factory
        return factory
    return deco


def collector_names() -> List[str]:
    return sorted(_REGISTRY)


def new_collector(name: str) -> MetricCollector:
    return _REGISTRY[name]()


@register_collector("local-cpu")
class LocalCpuCollector(MetricCollector):
    name = "local-cpu"

    def collect(self) -> Dict[str, float]:
        try:
            import psutil
            return {"cpu_pct": float(psutil.cpu_percent(interval=None)),
                    "cpu_count": float(psutil.cpu_count() or 1)}
        except Exception:
            return {"cpu_pct": 0.0, "cpu_count": 1.0}


@register_collector("local-memory")
class LocalMemoryCollector(MetricCollector):
    name = "local-memory"

    def collect(self) -> Dict[str, float]:
        try:
            import psutil
            vm = psutil.virtual_memory()
            return {"memory_pct": float(vm.percent),
                    "memory_total": float(vm.total),
                    "memory_used": float(vm.used)}
        except Exception:
            return {"memory_pct": 0.0, "memory_total": 0.0,
                    "memory_used": 0.0}


class MetricCollectManager:
    """Runs registered collectors, keeps a bounded sample window."""

    def __init__(self, names: Optional[List[str]] = None,
                 window: int = 600):
        names = names if names is not None else collector_names()
        self.collectors = [new_collector(n) for n in names]
        self.samples: Deque[Tuple[float, Dict[str, float]]] = \
            deque(maxlen=window)

    def collect_once(self, now: Optional[float] = None) -> Dict[str, float]:
        merged: Dict[str, float] = {}
        for c in self.collectors:
            merged.update(c.collect())
        self.samples.append((now if now is not None else time.time(),
                             merged))
        return merged

    def inject(self, sample: Dict[str, float],
               now: Optional[float] = None) -> None:
        """Test hook: record a synthetic sample."""
        self.samples.append((now if now is not None else time.time(),
                             dict(sample)))


class ResourceUsageGetter:
    """resource_usage_getter.go:35: windowed average usage the
    oversubscription policy consumes (5-min avg by default — the same
    horizon the scheduler's usage plugin assumes)."""

    def __init__(self, manager: MetricCollectManager,
                 window_seconds: float = 300.0):
        self.manager = manager
        self.window = window_seconds

    def usage(self, metric: str, now: Optional[float] = None) -> float:
        now = now if now is not None else time.time()
        vals = [s[metric] for (ts, s) in self.manager.samples
                if now - ts <= self.window and metric in s]
        return sum(vals) / len(vals) if vals else 0.0

    def cpu_pct(self, now: Optional[float] = None) -> float:
        return self.usage("cpu_pct", now)

    def memory_pct(self, now: Optional[float] = None) -> float:
        return self.usage("memory_pct", now)
