"""ColocationConfig controller (reference ``pkg/controllers/
colocationconfig``): projects ColocationConfig CRDs onto matching nodes
as effective-config annotations the node agents consume."""

from __future__ import annotations

import json

from .framework import Controller, register

ANN_EFFECTIVE = "colocation.volcano.sh/effective-config"


@register("colocationconfig")
class ColocationConfigController(Controller):
    watch_kinds = ("ColocationConfig", "Node")

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty = True

    def handle(self, ev) -> None:
        self._dirty = True

    def resync(self) -> None:
        if not self._dirty:
            return
        self._dirty = False
        configs = self.store.list("ColocationConfig")
        for node in self.store.list("Node"):
            effective = None
            for cfg in configs:
                if all(node.meta.labels.get(k) == v
                       for k, v in cfg.node_selector.items()):
                    effective = {
                        "cpuBurst": cfg.cpu_burst_enable,
                        "memoryQoS": cfg.memory_qos_enable,
                        "oversubscription": cfg.oversubscription_enable,
                        "oversubscriptionRatio": cfg.oversubscription_ratio,
                        "networkQoS": cfg.network_qos_enable,
                        "offlineBandwidthShare": cfg.offline_bandwidth_share,
                    }
            val = json.dumps(effective, sort_keys=True) if effective else None
            cur = node.meta.annotations.get(ANN_EFFECTIVE)
            if val != cur:
                if val is None:
                    node.meta.annotations.pop(ANN_EFFECTIVE, None)
                else:
                    node.meta.annotations[ANN_EFFECTIVE] = val
                self.store.update("Node", node)
