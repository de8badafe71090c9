"""HyperNode tree — recursive network-topology domains.

Reference: ``topology/v1alpha1`` HyperNode CRD (tier + member selectors,
hypernode_types.go:61-106) and the scheduler-side tree with LCA ancestry
(``pkg/scheduler/api/hyper_node_info.go``).  Lower tier = closer (one
leaf switch); higher tier = wider domain.
"""

from __future__ import annotations

import re
from typing import Dict, List, Optional, Set

from .objects import HyperNode


class HyperNodeTree:
    def __init__(self, hypernodes: List[HyperNode], node_names: List[str],
                 node_labels: Optional[Dict[str, Dict[str, str]]] = None):
        self.by_name: Dict[str, HyperNode] = {h.meta.name: h
                                              for h in hypernodes}
        self.parent: Dict[str, str] = {}
        self.members: Dict[str, Set[str]] = {}    # hypernode → node names
        self.tier: Dict[str, int] = {h.meta.name: h.tier for h in hypernodes}
        node_set = list(node_names)
        labels = node_labels or {}

        def select(selector, pool: List[str]) -> List[str]:
            """member_selector semantics (reference pkg/util/
            member_selector.go: exact / regex / label matching)."""
            out: List[str] = []
            pool_set = set(pool)
            if selector.exact_match:
                out.extend(n for n in selector.exact_match if n in pool_set)
            if selector.regex_match:
                rx = re.compile(selector.regex_match)
                out.extend(n for n in pool if rx.fullmatch(n))
            if selector.label_match:
                out.extend(n for n in pool
                           if all(labels.get(n, {}).get(k) == v
                                  for k, v in selector.label_match.items()))
            return out

        # resolve bottom-up by tier
        for h in sorted(hypernodes, key=lambda x: x.tier):
            nodes: Set[str] = set()
            for m in h.members:
                if m.type == "Node":
                    nodes.update(select(m.selector, node_set))
                else:
                    for child in select(m.selector,
                                        list(self.by_name)):
                        self.parent[child] = h.meta.name
                        nodes.update(self.members.get(child, set()))
            self.members[h.meta.name] = nodes

        # leaf (lowest-tier) hypernode per node
        self.leaf_of: Dict[str, str] = {}
        for name, nodes in self.members.items():
            for n in nodes:
                cur = self.leaf_of.get(n)
                if cur is None or self.tier[name] < self.tier[cur]:
                    self.leaf_of[n] = name

    def ancestors(self, hname: str) -> List[str]:
        out = [hname]
        while hname in self.parent:
            hname = self.parent[hname]
            out.append(hname)
        return out

    def lca_tier(self, node_a: str, node_b: str) -> Optional[int]:
        """Tier of the closest common domain of two nodes (reference
        hyper_node_info.go LCA ancestry)."""
        la, lb = self.leaf_of.get(node_a), self.leaf_of.get(node_b)
        if la is None or lb is None:
            return None
        aa = self.ancestors(la)
        ab = set(self.ancestors(lb))
        for h in aa:
            if h in ab:
                return self.tier[h]
        return None

    def domains_by_tier(self, max_tier: Optional[int] = None) -> List[str]:
        names = [n for n in self.members
                 if max_tier is None or self.tier[n] <= max_tier]
        return sorted(names, key=lambda n: (self.tier[n], n))
