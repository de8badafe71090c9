"""Conformance plugin (reference ``plugins/conformance``): never evict
kube-system / critical pods."""

from __future__ import annotations

from .base import Plugin, register

CRITICAL_PRIORITY = 2_000_000_000   # system-cluster-critical threshold


@register("conformance")
class ConformancePlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        def evictable(evictor, candidates):
            out = []
            for v in candidates:
                if v.namespace == "kube-system":
                    continue
                if v.priority >= CRITICAL_PRIORITY:
                    continue
                out.append(v)
            return out

        ssn.preemptable_fns.append(evictable)
        ssn.reclaimable_fns.append(evictable)
