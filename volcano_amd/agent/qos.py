"""QoS handlers (reference ``pkg/agent/events/handlers/{cpuqos,
memoryqos,memoryqosv2}`` and ``pkg/networkqos`` — cgroup v1/v2 knobs and
the tc/eBPF bandwidth limiter).

This environment has no cgroup tree or tc netlink to drive, so the
handlers compute the SAME control values the reference writes
(cpu.max/cpu burst quota, memory.high/low, online/offline bandwidth
watermarks) and publish them as pod/node annotations — the enforcement
contract an on-node enforcer (or test) reads.  QoS classes follow the
reference's LC/HLS/LS/BE model (docs/design/colocation/Overview.md):
offline (preemptable/best-effort) pods get throttled first.
"""

from __future__ import annotations

from ..api.objects import ANN_PREEMPTABLE

ANN_CPU_QUOTA = "qos.volcano.sh/cpu-quota-milli"
ANN_MEM_HIGH = "qos.volcano.sh/memory-high"
ANN_NET_LIMIT = "qos.volcano.sh/offline-bandwidth-bps"


def _offline(pod) -> bool:
    return pod.meta.annotations.get(ANN_PREEMPTABLE) == "true" or \
        pod.best_effort


class CpuQosHandler:
    """Offline pods' cpu quota shrinks as node pressure rises."""

    def __init__(self, low: float = 50.0, high: float = 80.0):
        self.low, self.high = low, high

    def handle(self, usage) -> None:
        for p in self.store.list(
                "Pod", selector=lambda p: p.node_name == self.node_name
                and p.phase == "Running" and _offline(p)):
            req = p.request.milli_cpu or 1000.0
            if usage.cpu_pct >= self.high:
                quota = req * 0.1
            elif usage.cpu_pct >= self.low:
                quota = req * 0.5
            else:
                quota = req     # full burst allowed
            val = str(int(quota))
            if p.meta.annotations.get(ANN_CPU_QUOTA) != val:
                p.meta.annotations[ANN_CPU_QUOTA] = val
                self.store.update("Pod", p)


class MemoryQosHandler:
    """memory.high for offline pods = request × factor (cgroup v2
    memoryqosv2 analog)."""

    def __init__(self, factor: float = 1.2):
        self.factor = factor

    def handle(self, usage) -> None:
        for p in self.store.list(
                "Pod", selector=lambda p: p.node_name == self.node_name
                and p.phase == "Running" and _offline(p)):
            val = str(int(p.request.memory * self.factor))
            if p.meta.annotations.get(ANN_MEM_HIGH) != val:
                p.meta.annotations[ANN_MEM_HIGH] = val
                self.store.update("Pod", p)


class NetworkQosHandler:
    """Online/offline bandwidth watermarks (reference pkg/networkqos:
    eBPF map values ONLINE_BANDWIDTH_WATERMARK etc.); offline share is
    squeezed when online traffic needs the headroom."""

    def __init__(self, total_bps: float = 25e9, offline_share: float = 0.3):
        self.total = total_bps
        self.offline_share = offline_share

    def handle(self, usage) -> None:
        node = self.store.get("Node", "default", self.node_name)
        if node is None:
            return
        # high node pressure ⇒ halve the offline bandwidth budget
        share = self.offline_share * (0.5 if usage.cpu_pct > 80 else 1.0)
        val = str(int(self.total * share))
        if node.meta.annotations.get(ANN_NET_LIMIT) != val:
            node.meta.annotations[ANN_NET_LIMIT] = val
            self.store.update("Node", node)
