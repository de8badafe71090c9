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
                 instance_label: str = "kubernetes.io/hostname",
                 transport=None):
        self.address = address.rstrip("/")
        self.timeout = timeout
        self.instance_label = instance_label
        # transport(url, params) -> parsed JSON; injectable for tests
        self.transport = transport

    def _get(self, url: str, params: dict):
        if self.transport is not None:
            return self.transport(url, params)
        import httpx
        r = httpx.get(url, params=params, timeout=self.timeout)
        r.raise_for_status()
        return r.json()

    def _query(self, expr: str) -> Optional[float]:
        try:
            data = self._get(f"{self.address}/api/v1/query",
                             {"query": expr})
            result = data["data"]["result"]
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


class ElasticsearchSource(MetricsSource):
    """reference metrics_client_elasticsearch.go:44-103: an avg
    aggregation over the last 5 minutes of cpu/memory utilization
    documents, one search per node (hostname-keyed index)."""

    def __init__(self, address: str, index: str = "metricbeat-*",
                 timeout: float = 5.0,
                 hostname_field: str = "host.hostname", transport=None):
        self.address = address.rstrip("/")
        self.index = index
        self.timeout = timeout
        self.hostname_field = hostname_field
        # transport(url, json_body) -> parsed JSON; injectable for tests
        self.transport = transport

    def _search(self, body: dict):
        if self.transport is not None:
            return self.transport(f"{self.address}/{self.index}/_search",
                                  body)
        import httpx
        r = httpx.post(f"{self.address}/{self.index}/_search", json=body,
                       timeout=self.timeout)
        r.raise_for_status()
        return r.json()

    def _avg(self, hostname: str, field: str) -> Optional[float]:
        body = {
            "size": 0,
            "query": {"bool": {"filter": [
                {"term": {self.hostname_field: hostname}},
                {"range": {"@timestamp": {"gte": "now-5m"}}},
            ]}},
            "aggs": {"avg_usage": {"avg": {"field": field}}},
        }
        try:
            data = self._search(body)
            v = data["aggregations"]["avg_usage"]["value"]
            return float(v) if v is not None else None
        except Exception:
            return None

    def node_usage(self, nodes) -> NodeUsage:
        out = NodeUsage()
        for node in nodes:
            host = node.meta.labels.get("kubernetes.io/hostname",
                                        node.meta.name)
            cpu = self._avg(host, "system.cpu.total.norm.pct")
            mem = self._avg(host, "system.memory.actual.used.pct")
            out[node.meta.name] = {
                "cpu": (cpu or 0.0) * 100.0,
                "memory": (mem or 0.0) * 100.0,
            }
        return out


class CustomMetricsSource(MetricsSource):
    """reference metrics_client_prometheus_adapt.go:39 — the
    custom.metrics.k8s.io API (prometheus-adapter): GET
    /apis/custom.metrics.k8s.io/v1beta2/nodes/<node>/<metric>."""

    CPU_METRIC = "node_cpu_usage_avg"
    MEM_METRIC = "node_memory_usage_avg"

    def __init__(self, address: str, timeout: float = 5.0, transport=None):
        self.address = address.rstrip("/")
        self.timeout = timeout
        self.transport = transport

    def _metric(self, node: str, metric: str) -> Optional[float]:
        url = (f"{self.address}/apis/custom.metrics.k8s.io/v1beta2/"
               f"nodes/{node}/{metric}")
        try:
            if self.transport is not None:
                data = self.transport(url, {})
            else:
                import httpx
                r = httpx.get(url, timeout=self.timeout)
                r.raise_for_status()
                data = r.json()
            items = data.get("items") or []
            if items:
                return float(items[0]["value"])
        except Exception:
            return None
        return None

    def node_usage(self, nodes) -> NodeUsage:
        out = NodeUsage()
        for node in nodes:
            cpu = self._metric(node.meta.name, self.CPU_METRIC)
            mem = self._metric(node.meta.name, self.MEM_METRIC)
            out[node.meta.name] = {"cpu": cpu or 0.0, "memory": mem or 0.0}
        return out


SOURCES = {
    "annotation": AnnotationSource,
    "prometheus": PrometheusSource,
    "elasticsearch": ElasticsearchSource,
    "custom": CustomMetricsSource,
}


def new_source(kind: str = "annotation", **kw) -> MetricsSource:
    return SOURCES[kind](**kw)
