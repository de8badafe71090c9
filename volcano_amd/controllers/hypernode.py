"""HyperNode controller (reference ``pkg/controllers/hypernode/``):
builds/updates HyperNode CRDs from network-topology *discovery*
providers.  Providers (reference discovery/{label,ufm,fake}):

* ``label`` — derives a two-tier tree from node labels
  (``topology.volcano.sh/rack`` → tier-1, ``topology.volcano.sh/spine``
  → tier-2); the provider shape real clusters use without an IB fabric
  manager.
* ``static`` — passes through operator-provided HyperNodes (an external
  source writes them; the controller just validates).
* ``fabric`` — consumes a fabric-manager port dump (the UFM discoverer
  analog, reference discovery/ufm/ufm.go:228 fetchUFMData +
  :267 buildHyperNodes): a JSON list of interface records
  ``{"system_name": <switch>, "tier": N, "node_description": <node>,
  "peer_node_name": <switch>}``.  Tier-1 records attach compute nodes
  to leaf switches; tier-2 records attach leaf switches to spines.  No
  fabric REST endpoint exists in this environment, so the dump arrives
  as a file (``fabric_file`` attr / ``VAMD_FABRIC_TOPOLOGY`` env).
"""

from __future__ import annotations

import json
import os
from typing import Dict, List

from ..api.objects import HyperNode, HyperNodeMember, MemberSelector, ObjectMeta
from .framework import Controller, register

LBL_RACK = "topology.volcano.sh/rack"
LBL_SPINE = "topology.volcano.sh/spine"


@register("hypernode")
class HyperNodeController(Controller):
    watch_kinds = ("Node",)

    fabric_file: str = ""

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty = True
        if not self.fabric_file:
            self.fabric_file = os.environ.get("VAMD_FABRIC_TOPOLOGY", "")
        self._fabric_mtime = 0.0

    def handle(self, ev) -> None:
        self._dirty = True

    def resync(self) -> None:
        self._resync_fabric()
        if not self._dirty:
            return
        self._dirty = False
        nodes = self.store.list("Node")
        racks: Dict[str, List[str]] = {}
        spines: Dict[str, List[str]] = {}
        for n in nodes:
            rack = n.meta.labels.get(LBL_RACK)
            if rack:
                racks.setdefault(rack, []).append(n.meta.name)
                spine = n.meta.labels.get(LBL_SPINE)
                if spine:
                    spines.setdefault(spine, []).append(f"rack-{rack}")
        for rack, members in sorted(racks.items()):
            self._apply(HyperNode(
                meta=ObjectMeta(name=f"rack-{rack}"), tier=1,
                members=[HyperNodeMember(
                    type="Node",
                    selector=MemberSelector(exact_match=sorted(members)))]))
        for spine, child_racks in sorted(spines.items()):
            self._apply(HyperNode(
                meta=ObjectMeta(name=f"spine-{spine}"), tier=2,
                members=[HyperNodeMember(
                    type="HyperNode",
                    selector=MemberSelector(
                        exact_match=sorted(set(child_racks))))]))

    def _resync_fabric(self) -> None:
        """Fabric-dump provider: (re)build HyperNodes from the interface
        records whenever the dump file changes (the UFM discoverer's
        periodic re-fetch, reference ufm.go:183 periodicDiscovery)."""
        path = self.fabric_file
        if not path or not os.path.exists(path):
            return
        mtime = os.path.getmtime(path)
        if mtime <= self._fabric_mtime:
            return
        self._fabric_mtime = mtime
        try:
            with open(path) as f:
                records = json.load(f)
        except (OSError, ValueError):
            return
        leafs: Dict[str, set] = {}      # leaf switch -> compute nodes
        spines: Dict[str, set] = {}     # spine switch -> leaf switches
        for r in records:
            if not isinstance(r, dict) or not r.get("system_name"):
                continue
            sw = str(r["system_name"])
            tier = int(r.get("tier", 1))
            if tier <= 1:
                member = r.get("node_description") or r.get("peer_node_name")
                if member:
                    leafs.setdefault(sw, set()).add(str(member))
            else:
                peer = r.get("peer_node_name") or r.get("node_description")
                if peer:
                    spines.setdefault(sw, set()).add(f"leaf-{peer}")
        for sw, members in sorted(leafs.items()):
            self._apply(HyperNode(
                meta=ObjectMeta(name=f"leaf-{sw}"), tier=1,
                members=[HyperNodeMember(
                    type="Node",
                    selector=MemberSelector(exact_match=sorted(members)))]))
        for sw, children in sorted(spines.items()):
            self._apply(HyperNode(
                meta=ObjectMeta(name=f"spine-{sw}"), tier=2,
                members=[HyperNodeMember(
                    type="HyperNode",
                    selector=MemberSelector(exact_match=sorted(children)))]))

    def _apply(self, hn: HyperNode) -> None:
        cur = self.store.get("HyperNode", hn.meta.namespace, hn.meta.name)
        if cur is None:
            self.store.create("HyperNode", hn)
        else:
            new_members = [(m.type, tuple(m.selector.exact_match))
                           for m in hn.members]
            old_members = [(m.type, tuple(m.selector.exact_match))
                           for m in cur.members]
            if new_members != old_members or cur.tier != hn.tier:
                cur.members = hn.members
                cur.tier = hn.tier
                self.store.update("HyperNode", cur)
