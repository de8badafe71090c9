"""Topology-spread plugin (reference: predicates wraps the k8s
podtopologyspread filter).

Simplified hard semantics: pods declaring
``affinity = {"topologySpread": {"group": G, "topologyKey": K,
"maxSkew": s}}`` must spread across the values of node label K so that
no value's member count exceeds the minimum count by more than
``maxSkew``.

Mapping: at session open the plugin counts placed members of G per
topology value and forbids (dynamic bit) the nodes of every value that
is already at ``min_count + maxSkew``.  Enforcement is cycle-granular:
the counts refresh per cycle, so a class of k instances placed in one
cycle can transiently overshoot inside the cycle — exact for the
pod-per-cycle arrival pattern, convergent otherwise (documented
approximation; the within-class least-requested spread keeps instances
apart in practice).
"""

from __future__ import annotations

from collections import defaultdict
from typing import Dict

from ...api.types import TaskStatus
from ..tensors import set_plane_bit
from .base import Plugin, register

OCCUPY = (TaskStatus.ALLOCATED, TaskStatus.BINDING, TaskStatus.BOUND,
          TaskStatus.RUNNING)


def _spec(pod):
    aff = pod.affinity if pod is not None else None
    if isinstance(aff, dict):
        s = aff.get("topologySpread")
        if isinstance(s, dict) and s.get("group") and s.get("topologyKey"):
            return s
    return None


@register("topologyspread")
class TopologySpreadPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors

        # group → topology value → placed member count
        counts: Dict[tuple, Dict[str, int]] = defaultdict(
            lambda: defaultdict(int))
        specs: Dict[tuple, dict] = {}
        for job in ssn.jobs.values():
            for t in job.tasks.values():
                s = _spec(t.pod)
                if s is None:
                    continue
                gk = (s["group"], s["topologyKey"])
                specs[gk] = s
                if t.status in OCCUPY and t.node_name in ssn.nodes:
                    ni = ssn.nodes[t.node_name]
                    val = ni.node.meta.labels.get(s["topologyKey"])
                    if val is not None:
                        counts[gk][val] += 1
        if not specs:
            return

        # nodes per topology value (per key)
        nodes_by_val: Dict[str, Dict[str, list]] = defaultdict(
            lambda: defaultdict(list))
        for ni in ssn.nodes.values():
            for (_, key) in specs:
                val = ni.node.meta.labels.get(key)
                if val is not None:
                    nodes_by_val[key][val].append(ni.node_id)

        bits: Dict[tuple, int] = {}
        for gk, s in specs.items():
            group, key = gk
            max_skew = int(s.get("maxSkew", 1))
            vals = nodes_by_val.get(key, {})
            if not vals:
                continue
            cnt = counts.get(gk, {})
            floor = min((cnt.get(v, 0) for v in vals), default=0)
            saturated = [v for v in vals
                         if cnt.get(v, 0) >= floor + max_skew]
            # nodes without the topology label are always out of bounds
            unlabeled = [ni.node_id for ni in ssn.nodes.values()
                         if ni.node.meta.labels.get(key) is None]
            ids = unlabeled + [nid for v in saturated for nid in vals[v]]
            if ids:
                bits[gk] = nt.add_dynamic_bit(
                    f"tsp:{group}:{key}:{floor}", sorted(set(ids)))

        def hook(tclass, job, require, forbid):
            s = _spec(tclass.tasks[0].pod)
            if s is None:
                return
            bit = bits.get((s["group"], s["topologyKey"]))
            if bit is not None:
                set_plane_bit(forbid, bit)

        ssn.class_constraint_hooks.append(hook)
