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
    """Offline pods' cpu quota shrinks as node pressure rises.  With a
    CgroupDriver the quota is written to the pod slice's cpu.max /
    cpu.cfs_quota_us (the reference's kernel contract); the annotation
    stays as the observable report."""

    def __init__(self, low: float = 50.0, high: float = 80.0,
                 cgroup=None):
        self.low, self.high = low, high
        self.cgroup = cgroup

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
                if self.cgroup is not None:
                    uid = p.meta.uid or p.meta.key
                    self.cgroup.set_cpu_quota(uid, True, quota)
                    self.cgroup.set_cpu_burst(uid, True,
                                              int(quota * 100))


class MemoryQosHandler:
    """memory.high for offline pods = request × factor (cgroup v2
    memoryqosv2 analog); with a CgroupDriver the value lands in
    memory.high (and guaranteed pods would get memory.low/min)."""

    def __init__(self, factor: float = 1.2, cgroup=None):
        self.factor = factor
        self.cgroup = cgroup

    def handle(self, usage) -> None:
        for p in self.store.list(
                "Pod", selector=lambda p: p.node_name == self.node_name
                and p.phase == "Running" and _offline(p)):
            val = str(int(p.request.memory * self.factor))
            if p.meta.annotations.get(ANN_MEM_HIGH) != val:
                p.meta.annotations[ANN_MEM_HIGH] = val
                self.store.update("Pod", p)
                if self.cgroup is not None:
                    uid = p.meta.uid or p.meta.key
                    self.cgroup.set_memory_high(
                        uid, True, p.request.memory * self.factor)


class NetworkQosHandler:
    """Online/offline bandwidth watermarks (reference pkg/networkqos:
    eBPF map values ONLINE_BANDWIDTH_WATERMARK etc.); offline share is
    squeezed when online traffic needs the headroom."""

    def __init__(self, total_bps: float = 25e9, offline_share: float = 0.3,
                 enforcer=None):
        self.total = total_bps
        self.offline_share = offline_share
        self.enforcer = enforcer      # NetQoSEnforcer (tc/eBPF layer)

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
        if self.enforcer is not None:
            if not self.enforcer.attached:
                self.enforcer.attach()
            # online pressure approximated from node cpu utilization
            online = int(self.enforcer.config.total_bps
                         * min(usage.cpu_pct, 100.0) / 100.0)
            self.enforcer.adjust(online)
