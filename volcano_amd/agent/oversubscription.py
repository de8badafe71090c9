"""Oversubscription handler (reference ``pkg/agent/oversubscription``):
computes the node's oversold CPU/memory from real utilization and
reports it via node annotations; the scheduler consumes it as extra
allocatable (NodeInfo.oversubscription, reference node_info.go:83-89).
"""

from __future__ import annotations

from ..api.resource import CPU, MEMORY

ANN_OVERSUB_CPU = "volcano.sh/oversubscription-cpu"
ANN_OVERSUB_MEM = "volcano.sh/oversubscription-memory"
ANN_OVERSUB_TYPES = "volcano.sh/oversubscription-types"


class OversubscriptionHandler:
    """Oversold amount = allocatable × (1 − usage%) × ratio."""

    def __init__(self, ratio: float = 0.6, high_watermark: float = 80.0):
        self.ratio = ratio
        self.high = high_watermark

    def handle(self, usage) -> None:
        node = self.store.get("Node", "default", self.node_name)
        if node is None:
            return
        if usage.cpu_pct >= self.high or usage.mem_pct >= self.high:
            cpu_over = mem_over = 0.0      # pressure: stop overselling
        else:
            cpu_over = node.allocatable.get(CPU) * \
                (1.0 - usage.cpu_pct / 100.0) * self.ratio
            mem_over = node.allocatable.get(MEMORY) * \
                (1.0 - usage.mem_pct / 100.0) * self.ratio
        ann = node.meta.annotations
        new = (str(int(cpu_over)), str(int(mem_over)))
        if (ann.get(ANN_OVERSUB_CPU), ann.get(ANN_OVERSUB_MEM)) != new:
            ann[ANN_OVERSUB_CPU], ann[ANN_OVERSUB_MEM] = new
            ann[ANN_OVERSUB_TYPES] = "cpu,memory"
            self.store.update("Node", node)
