"""Prometheus-real metrics registry (reference ``pkg/scheduler/metrics/
metrics.go:57-191``): the same metric names, HistogramVec semantics and
exponential bucket ranges as the reference, exposed in the Prometheus
text format, while keeping the zero-dependency hot-path API
(``observe``/``inc`` with ``name[:label[:label]]`` keys).

Internal keys map to reference metrics at export:

* ``e2e_scheduling_latency``            → volcano_e2e_scheduling_latency_milliseconds
* ``open_session_duration``             → volcano_open_session_duration_milliseconds
* ``action_scheduling_latency:<a>``     → volcano_action_scheduling_latency_milliseconds{action="a"}
* ``plugin_scheduling_latency:<p>:<e>`` → volcano_plugin_scheduling_latency_milliseconds{plugin="p",OnSession="e"}
* ``task_scheduling_latency``           → volcano_task_scheduling_latency_milliseconds
* ``scheduling_stage_duration:<s>``     → volcano_scheduling_stage_duration_milliseconds{stage="s"}
* counters keep their name (``schedule_attempts_total:<result>`` →
  volcano_schedule_attempts_total{result="r"}, ...)

Values passed to ``observe`` are SECONDS (the call sites time with
perf_counter); export converts to milliseconds to match the reference
units.
"""

from __future__ import annotations

import bisect
import math
import threading
from collections import defaultdict
from typing import Dict, List, Optional, Tuple


def exponential_buckets_range(lo: float, hi: float, count: int) -> List[float]:
    """prometheus.ExponentialBucketsRange (client_golang): count buckets,
    first upper bound lo, last hi, exponentially spaced."""
    if count < 1:
        return [hi]
    growth = (hi / lo) ** (1.0 / max(count - 1, 1))
    return [lo * (growth ** i) for i in range(count)]


# (prom name, label key, buckets in ms) per internal prefix —
# bucket ranges mirror metrics.go:57-146
_HISTO_SPEC: Dict[str, Tuple[str, Optional[Tuple[str, ...]], List[float]]] = {
    "e2e_scheduling_latency": (
        "volcano_e2e_scheduling_latency_milliseconds", None,
        exponential_buckets_range(1, 5000, 20)),
    "open_session_duration": (
        "volcano_open_session_duration_milliseconds", None,
        exponential_buckets_range(1, 5000, 20)),
    "action_scheduling_latency": (
        "volcano_action_scheduling_latency_milliseconds", ("action",),
        exponential_buckets_range(1, 2000, 20)),
    "plugin_scheduling_latency": (
        "volcano_plugin_scheduling_latency_milliseconds",
        ("plugin", "OnSession"),
        exponential_buckets_range(0.1, 100, 15)),
    "task_scheduling_latency": (
        "volcano_task_scheduling_latency_milliseconds", None,
        exponential_buckets_range(50, 60000, 30)),
    "scheduling_stage_duration": (
        "volcano_scheduling_stage_duration_milliseconds", ("stage",),
        exponential_buckets_range(0.1, 500, 20)),
    # repo-specific stages ride the same histogram machinery
    "allocate": (
        "volcano_scheduling_stage_duration_milliseconds", ("stage",),
        exponential_buckets_range(0.1, 500, 20)),
}

_COUNTER_LABEL: Dict[str, Optional[str]] = {
    "schedule_attempts_total": "result",
    "eviction_transactions_total": "reason",
    "unschedule_task_count": "job_id",
}


class _Histo:
    __slots__ = ("buckets", "counts", "total", "count")

    def __init__(self, buckets: List[float]):
        self.buckets = buckets          # upper bounds, ms
        self.counts = [0] * (len(buckets) + 1)   # +Inf tail
        self.total = 0.0
        self.count = 0

    def observe_ms(self, v: float) -> None:
        self.counts[bisect.bisect_left(self.buckets, v)] += 1
        self.total += v
        self.count += 1


class MetricsRegistry:
    def __init__(self):
        self._lock = threading.Lock()
        self._observations: Dict[str, List[float]] = defaultdict(list)
        self._histos: Dict[Tuple[str, Tuple[str, ...]], _Histo] = {}
        self._counters: Dict[str, float] = defaultdict(float)
        self._gauges: Dict[str, float] = {}

    # -- hot-path API --------------------------------------------------------
    def observe(self, name: str, value: float) -> None:
        """value in SECONDS; `name` may carry :labels."""
        base, _, _ = name.partition(":")
        with self._lock:
            obs = self._observations[name]
            obs.append(value)
            if len(obs) > 10000:
                del obs[: len(obs) // 2]
            spec = _HISTO_SPEC.get(base)
            if spec is not None:
                parts = tuple(name.split(":")[1:])
                h = self._histos.get((base, parts))
                if h is None:
                    h = self._histos[(base, parts)] = _Histo(spec[2])
                h.observe_ms(value * 1000.0)

    def inc(self, name: str, value: float = 1.0) -> None:
        with self._lock:
            self._counters[name] += value

    def set_gauge(self, name: str, value: float) -> None:
        with self._lock:
            self._gauges[name] = value

    # -- summaries (bench / tests) ------------------------------------------
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

    # -- exposition ----------------------------------------------------------
    @staticmethod
    def _fmt(v: float) -> str:
        if v == math.inf:
            return "+Inf"
        return repr(round(v, 6))

    def export_prometheus(self) -> str:
        """Prometheus text exposition format 0.0.4: HELP/TYPE headers,
        histogram bucket/sum/count series with le labels — parseable by
        promtool and scrapeable as the reference's endpoint is."""
        out: List[str] = []
        with self._lock:
            histos = dict(self._histos)
            counters = dict(self._counters)
            gauges = dict(self._gauges)
        # histograms grouped by prom name
        by_prom: Dict[str, List[Tuple[Tuple[str, ...], _Histo]]] = \
            defaultdict(list)
        for (base, labels), h in sorted(histos.items()):
            prom, lkeys, _ = _HISTO_SPEC[base]
            by_prom[prom].append((self._labelset(lkeys, labels), h))
        for prom in sorted(by_prom):
            out.append(f"# HELP {prom} (reference metrics.go parity)")
            out.append(f"# TYPE {prom} histogram")
            for lbl, h in by_prom[prom]:
                cum = 0
                for ub, c in zip(h.buckets + [math.inf],
                                 h.counts):
                    cum += c
                    le = self._merge_labels(lbl, ("le", self._fmt(ub)))
                    out.append(f"{prom}_bucket{le} {cum}")
                out.append(f"{prom}_sum{self._render(lbl)} "
                           f"{self._fmt(h.total)}")
                out.append(f"{prom}_count{self._render(lbl)} {h.count}")
        for name in sorted(counters):
            base, _, label = name.partition(":")
            prom = base if base.startswith("volcano_") \
                else f"volcano_{base}".replace(":", "_")
            lkey = _COUNTER_LABEL.get(base)
            lbl = ((lkey, label),) if (lkey and label) else ()
            if not out or f"# TYPE {prom} counter" not in out:
                out.append(f"# HELP {prom} counter")
                out.append(f"# TYPE {prom} counter")
            out.append(f"{prom}{self._render(lbl)} "
                       f"{self._fmt(counters[name])}")
        seen_gauge = set()
        for name in sorted(gauges):
            prom = name if name.startswith("volcano_") else f"volcano_{name}"
            base = prom.split("{", 1)[0]
            if base not in seen_gauge:
                seen_gauge.add(base)
                out.append(f"# HELP {base} gauge")
                out.append(f"# TYPE {base} gauge")
            out.append(f"{prom} {self._fmt(gauges[name])}")
        return "\n".join(out) + "\n"

    @staticmethod
    def _labelset(lkeys, labels) -> Tuple[Tuple[str, str], ...]:
        if not lkeys:
            return ()
        return tuple(zip(lkeys, labels + ("",) * (len(lkeys) - len(labels))))

    @staticmethod
    def _render(lbl: Tuple[Tuple[str, str], ...]) -> str:
        if not lbl:
            return ""
        body = ",".join(f'{k}="{v}"' for k, v in lbl)
        return "{" + body + "}"

    @classmethod
    def _merge_labels(cls, lbl, extra) -> str:
        return cls._render(tuple(lbl) + (extra,))

    def export_text(self) -> str:
        """Legacy flat summary export (SIGUSR1 dump, tests)."""
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

    def load_rocprof_stats(self, rows: Dict[str, float]) -> None:
        """Fold per-kernel GPU time (rocprofv3 --stats output, kernel →
        total microseconds) into the same exposition as gauges:
        volcano_kernel_time_microseconds{kernel="..."}."""
        with self._lock:
            for kernel, us in rows.items():
                safe = kernel.replace('"', "").replace("\\", "")[:120]
                self._gauges[
                    f'volcano_kernel_time_microseconds{{kernel="{safe}"}}'] \
                    = float(us)

    def reset(self) -> None:
        with self._lock:
            self._observations.clear()
            self._counters.clear()
            self._histos.clear()
            self._gauges.clear()


METRICS = MetricsRegistry()
