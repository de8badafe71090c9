"""Shard assignment across scheduler ranks.

Reference: the sharding controller + NodeShard CRD partition cluster
nodes across scheduler instances with a 2-phase handoff
(``pkg/controllers/sharding/``, ``shard/v1alpha1``) —
"hard" mode: a scheduler only places pods on nodes it owns
(``util/predicate_helper.go:103 GetPredicatedNodeByShard``).

The MI355X deployment shape is one scheduler rank per GPU on one node:
rank r owns nodes round-robin (balanced capacity) and jobs by stable
hash (balanced load); ownership is deterministic from (rank, world), so
no coordination traffic is needed until membership changes — the
NodeShard CRD records the assignment for observability and handoff.
"""

from __future__ import annotations

import zlib
from typing import List, Optional

from ..api.objects import NodeShard, ObjectMeta


def stable_hash(s: str) -> int:
    return zlib.crc32(s.encode())


class ShardingPolicy:
    def __init__(self, rank: int, world: int):
        assert 0 <= rank < world
        self.rank = rank
        self.world = world

    def owns_node_index(self, i: int) -> bool:
        return i % self.world == self.rank

    def owns_node(self, name: str, index: Optional[int] = None) -> bool:
        if index is not None:
            return self.owns_node_index(index)
        return stable_hash(name) % self.world == self.rank

    def owns_job(self, key: str) -> bool:
        return stable_hash(key) % self.world == self.rank

    def filter_nodes(self, names: List[str]) -> List[str]:
        return [n for i, n in enumerate(sorted(names))
                if self.owns_node_index(i)]

    def to_nodeshard(self, all_nodes: List[str]) -> NodeShard:
        owned = self.filter_nodes(all_nodes)
        return NodeShard(
            meta=ObjectMeta(name=f"shard-{self.rank}"),
            nodes_desired=owned, nodes_in_use=owned)
