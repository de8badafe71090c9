"""Inter-pod affinity plugin (reference: predicates wraps the k8s
interpodaffinity filter, plugins/predicates/predicates.go:34-47).

Simplified hard semantics at hostname topology, expressed through the
existing kernel machinery — no per-(pod,pod) matching in the hot loop:

* **anti-affinity** (``affinity = {"podAntiAffinity": {"group": G}}``):
  handled structurally, not by this plugin — group G is a synthetic unit
  resource dim (``paa:G``): every node offers 1 (tensors.py pack), every
  member requests 1 (TaskInfo.from_pod), so the score/cap kernel's
  ordinary capacity math enforces at-most-one-member-per-node.
* **affinity** (``{"podAffinity": {"group": G}}``): the pod must land on
  a node already hosting a member of G — a dynamic require bit over the
  hosting nodes (vacuously unconstrained while the group has no placed
  member: the first member anchors).
"""

from __future__ import annotations

from typing import Dict, Set

from ...api.types import TaskStatus
from ..tensors import set_plane_bit
from .base import Plugin, register

OCCUPY = (TaskStatus.ALLOCATED, TaskStatus.BINDING, TaskStatus.BOUND,
          TaskStatus.RUNNING)


def _group(pod, kind: str):
    aff = pod.affinity if pod is not None else None
    if not isinstance(aff, dict):
        return None
    spec = aff.get(kind)
    if isinstance(spec, dict):
        return spec.get("group")
    return None


@register("interpodaffinity")
class InterPodAffinityPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        affin_hosts: Dict[str, Set[int]] = {}
        for job in ssn.jobs.values():
            for t in job.tasks.values():
                g = _group(t.pod, "podAffinity")
                if g and t.status in OCCUPY and t.node_name in ssn.nodes:
                    affin_hosts.setdefault(g, set()).add(
                        ssn.nodes[t.node_name].node_id)
        if not affin_hosts:
            return
        bits = {g: nt.add_dynamic_bit(f"paff:{g}", sorted(ids))
                for g, ids in affin_hosts.items()}

        def hook(tclass, job, require, forbid):
            g = _group(tclass.tasks[0].pod, "podAffinity")
            if g and g in bits:
                set_plane_bit(require, bits[g])

        ssn.class_constraint_hooks.append(hook)
