"""Node agent event manager (reference ``pkg/agent/events/eventsmgr.go``:
a DaemonSet per node — probes feed handlers; handlers enforce colocation
QoS: cpuburst/cpuqos/memoryqos/eviction/networkqos/oversubscription).

One EventsManager instance represents the agent on one node: probes
produce ``NodeUsage`` samples (here from node/pod state in the store —
the cadvisor/metriccollect analog), handlers react.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import List, Optional

from ..store import ObjectStore


@dataclass
class NodeUsage:
    node_name: str
    cpu_pct: float = 0.0
    mem_pct: float = 0.0
    pod_count: int = 0


class UsageProbe:
    """reference agent/events/probes/noderesources: samples utilization.

    Source: explicit injection (tests) or the node's usage annotations
    (volcano.sh/cpu-usage, volcano.sh/memory-usage — same keys the usage
    plugin and metrics pipeline use)."""

    def __init__(self, store: ObjectStore, node_name: str,
                 usage_getter=None):
        self.store = store
        self.node_name = node_name
        self.injected: Optional[NodeUsage] = None
        # metriccollect-backed source (ResourceUsageGetter): real local
        # counters when running as a node daemon
        self.usage_getter = usage_getter

    def sample(self) -> Optional[NodeUsage]:
        if self.injected is not None:
            return self.injected
        if self.usage_getter is not None:
            pods = self.store.list(
                "Pod", selector=lambda p: p.node_name == self.node_name
                and p.phase in ("Running", "Pending"))
            return NodeUsage(node_name=self.node_name,
                             cpu_pct=self.usage_getter.cpu_pct(),
                             mem_pct=self.usage_getter.memory_pct(),
                             pod_count=len(pods))
        node = self.store.get("Node", "default", self.node_name)
        if node is None:
            return None
        ann = node.meta.annotations
        pods = self.store.list(
            "Pod", selector=lambda p: p.node_name == self.node_name
            and p.phase in ("Running", "Pending"))
        return NodeUsage(
            node_name=self.node_name,
            cpu_pct=float(ann.get("volcano.sh/cpu-usage", 0.0)),
            mem_pct=float(ann.get("volcano.sh/memory-usage", 0.0)),
            pod_count=len(pods))


class EventsManager:
    def __init__(self, store: ObjectStore, node_name: str):
        self.store = store
        self.node_name = node_name
        self.probe = UsageProbe(store, node_name)
        self.handlers: List[object] = []

    def register(self, handler) -> None:
        handler.store = self.store
        handler.node_name = self.node_name
        self.handlers.append(handler)

    def tick(self) -> None:
        usage = self.probe.sample()
        if usage is None:
            return
        for h in self.handlers:
            h.handle(usage)
