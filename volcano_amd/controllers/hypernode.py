"""HyperNode controller (reference ``pkg/controllers/hypernode/``):
builds/updates HyperNode CRDs from network-topology *discovery*
providers.  Providers (reference discovery/{label,ufm,fake}):

* ``label`` — derives a two-tier tree from node labels
  (``topology.volcano.sh/rack`` → tier-1, ``topology.volcano.sh/spine``
  → tier-2); the provider shape real clusters use without an IB fabric
  manager.
* ``static`` — passes through operator-provided HyperNodes (the UFM/fake
  analog: an external source writes them; the controller just validates).
"""

from __future__ import annotations

from typing import Dict, List

from ..api.objects import HyperNode, HyperNodeMember, MemberSelector, ObjectMeta
from .framework import Controller, register

LBL_RACK = "topology.volcano.sh/rack"
LBL_SPINE = "topology.volcano.sh/spine"


@register("hypernode")
class HyperNodeController(Controller):
    watch_kinds = ("Node",)

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty = True

    def handle(self, ev) -> None:
        self._dirty = True

    def resync(self) -> None:
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
