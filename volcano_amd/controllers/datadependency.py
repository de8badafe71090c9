"""DataSourceClaim binder (reference ``staging/.../datadependency/
v1alpha1/types.go:32-201``): binds each claim to the DataSource whose
(system, type, name) triple matches, mirrors the bind set into the
source's status (claimRefs / boundClaims), marks claims with no match
``Lost``-free (they stay Pending), and applies the source's
reclaimPolicy — ``Delete`` removes a DataSource when its last bound
claim goes away."""

from __future__ import annotations

from ..store import EventType
from .framework import Controller, register


@register("datadependency")
class DataDependencyController(Controller):
    watch_kinds = ("DataSource", "DataSourceClaim")

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty = True

    def handle(self, ev) -> None:
        self._dirty = True
        if ev.kind == "DataSourceClaim" and ev.type == EventType.DELETED:
            self._dirty = True

    def resync(self) -> None:
        if not self._dirty:
            return
        self._dirty = False
        store = self.store
        sources = list(store.list("DataSource"))
        by_triple = {}
        for s in sources:
            by_triple.setdefault((s.system, s.type, s.name), s)

        bound: dict = {s.meta.name: [] for s in sources}
        for c in store.list("DataSourceClaim"):
            src = by_triple.get(
                (c.system, c.data_source_type, c.data_source_name))
            if src is None:
                if c.phase != "Pending" or c.bound_data_source:
                    c.phase = "Pending"
                    c.bound_data_source = ""
                    store.update("DataSourceClaim", c)
                continue
            bound[src.meta.name].append(c.meta.key)
            if c.phase != "Bound" or c.bound_data_source != src.meta.name:
                c.phase = "Bound"
                c.bound_data_source = src.meta.name
                store.update("DataSourceClaim", c)

        for s in sources:
            refs = sorted(bound.get(s.meta.name, []))
            had = bool(s.claim_refs)
            if s.claim_refs != refs or s.bound_claims != len(refs):
                s.claim_refs = refs
                s.bound_claims = len(refs)
                store.update("DataSource", s)
            if had and not refs and s.reclaim_policy == "Delete":
                # last bound claim released an ephemeral source
                store.delete("DataSource", s.meta.namespace, s.meta.name)
