"""Sharding controller (reference ``pkg/controllers/sharding/``,
SURVEY §2.4): partitions cluster nodes into NodeShard CRDs, one per
scheduler rank, with the 2-phase handoff (desired → toAdd/toRemove →
inUse) so two schedulers never own one node simultaneously
(shard/v1alpha1 types.go:33-71).
"""

from __future__ import annotations

from typing import List

from ..api.objects import NodeShard, ObjectMeta
from ..parallel.sharding import ShardingPolicy
from .framework import Controller, register


@register("sharding")
class ShardingController(Controller):
    watch_kinds = ("Node", "NodeShard")

    def __init__(self, shards: int = 1):
        self.shards = shards

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty = True

    def handle(self, ev) -> None:
        if ev.kind == "Node":
            self._dirty = True

    def resync(self) -> None:
        if not self._dirty:
            return
        self._dirty = False
        nodes = sorted(n.meta.name for n in self.store.list("Node"))
        for rank in range(self.shards):
            policy = ShardingPolicy(rank, self.shards)
            desired = policy.filter_nodes(nodes)
            name = f"shard-{rank}"
            cur = self.store.get("NodeShard", "default", name)
            if cur is None:
                # phase 1: publish desired; the owning scheduler confirms
                self.store.create("NodeShard", NodeShard(
                    meta=ObjectMeta(name=name),
                    nodes_desired=desired,
                    nodes_to_add=desired))
                continue
            if cur.nodes_desired != desired:
                in_use = set(cur.nodes_in_use)
                want = set(desired)
                cur.nodes_desired = desired
                cur.nodes_to_add = sorted(want - in_use)
                cur.nodes_to_remove = sorted(in_use - want)
                self.store.update("NodeShard", cur)

    # -- scheduler-side acknowledgement (cache/shard_coordinator.go) -------
    @staticmethod
    def acknowledge(store, rank: int) -> List[str]:
        """Phase 2: the scheduler adopts toAdd / releases toRemove and
        confirms inUse.  Returns the confirmed node list."""
        name = f"shard-{rank}"
        shard = store.get("NodeShard", "default", name)
        if shard is None:
            return []
        in_use = set(shard.nodes_in_use)
        in_use |= set(shard.nodes_to_add)
        in_use -= set(shard.nodes_to_remove)
        changed = sorted(in_use) != shard.nodes_in_use or \
            shard.nodes_to_add or shard.nodes_to_remove
        shard.nodes_in_use = sorted(in_use)
        shard.nodes_to_add = []
        shard.nodes_to_remove = []
        if changed:
            store.update("NodeShard", shard)
        return shard.nodes_in_use
