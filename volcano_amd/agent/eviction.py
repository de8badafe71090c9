"""Pressure eviction handler (reference ``pkg/agent/events/handlers/
eviction`` + oversubscription/policy): when node utilization crosses the
eviction watermark, evict best-effort/preemptable (offline) pods first
until below the low watermark."""

from __future__ import annotations

from ..api.objects import ANN_PREEMPTABLE


class EvictionHandler:
    def __init__(self, high_watermark: float = 90.0):
        self.high = high_watermark

    def handle(self, usage) -> None:
        if usage.cpu_pct < self.high and usage.mem_pct < self.high:
            return
        pods = self.store.list(
            "Pod", selector=lambda p: p.node_name == self.node_name
            and p.phase == "Running")
        # offline (preemptable / best-effort) first, largest first
        offline = [p for p in pods
                   if p.meta.annotations.get(ANN_PREEMPTABLE) == "true"
                   or p.best_effort]
        offline.sort(key=lambda p: -p.request.milli_cpu)
        for p in offline[:1]:            # one per tick; re-evaluate next
            p.phase = "Failed"
            p.meta.annotations["volcano.sh/evicted"] = "node-pressure"
            self.store.update("Pod", p)
