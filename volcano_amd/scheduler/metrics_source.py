"""Node real-usage metrics sources (reference ``pkg/scheduler/metrics/
source``: MetricsClient with prometheus / elasticsearch / custom-adapter
implementations feeding cache.setMetricsData).

The usage plugin and the node agent consume a ``MetricsSource``; the
default reads the node annotations the agent publishes; the Prometheus
source scrapes a real endpoint (``avg cpu/mem utilization over 5m``,
matching the reference's query shape).
"""

from __future__ import annotations

from typing import Optional

ANN_CPU_USAGE = "volcano.sh/cpu-usage"
ANN_MEM_USAGE = "volcano.sh/memory-usage"


class NodeUsage(dict):
    """{node_name: {"cpu": pct, "memory": pct}}"""


class MetricsSource:
    def node_usage(self, nodes) -> NodeUsage:  # pragma: no cover
        raise NotImplementedError


class AnnotationSource(MetricsSource):
    """Default: the node agent publishes usage as annotations."""

    def node_usage(self, nodes) -> NodeUsage:
        out = NodeUsage()
        for node in nodes:
            ann = node.meta.annotations
            out[node.meta.name] = {
                "cpu": float(ann.get(ANN_CPU_USAGE, 0.0)),
                "memory": float(ann.get(ANN_MEM_USAGE, 0.0)),
            }
        return out


class PrometheusSource(MetricsSource):
    """reference metrics_client_prometheus.go:35-97 — queries
    100*(1-avg(irate(node_cpu_seconds_total{mode='idle'}[5m]))) style
    expressions per node instance."""

    CPU_QUERY = ('100 - avg(rate(node_cpu_seconds_total{{mode="idle",'
                 'instance="{instance}"}}[5m])) * 100')
    MEM_QUERY = ('100 * (1 - node_memory_MemAvailable_bytes{{'
                 'instance="{instance}"}} / node_memory_MemTotal_bytes{{'
                 'instance="{instance}"}})')

    def __init__(self, address: str, timeout: float = 5.0,
                 instance_label: str = "kubernetes.io/hostname"):
        self.address = address.rstrip("/")
        self.timeout = timeout
        self.instance_label = instance_label

    def _query(self, expr: str) -> Optional[float]:
        import httpx
        try:
            r = httpx.get(f"{self.address}/api/v1/query",
                          params={"query": expr}, timeout=self.timeout)
            r.raise_for_status()
            result = r.json()["data"]["result"]
            if result:
                return float(result[0]["value"][1])
        except Exception:
            return None
        return None

    def node_usage(self, nodes) -> NodeUsage:
        out = NodeUsage()
        for node in nodes:
            inst = node.meta.labels.get(self.instance_label, node.meta.name)
            cpu = self._query(self.CPU_QUERY.format(instance=inst))
            mem = self._query(self.MEM_QUERY.format(instance=inst))
            out[node.meta.name] = {"cpu": cpu or 0.0, "memory": mem or 0.0}
        return out


SOURCES = {
    "annotation": AnnotationSource,
    "prometheus": PrometheusSource,
}


def new_source(kind: str = "annotation", **kw) -> MetricsSource:
    return SOURCES[kind](**kw)
