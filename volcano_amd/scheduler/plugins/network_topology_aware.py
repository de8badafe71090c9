"""Network-topology-aware plugin (reference
``plugins/network-topology-aware`` + the allocate action's hypernode
gradient search; design "Network Topology Aware
Scheduling").

A job whose PodGroup sets ``networkTopology`` ({"mode": "hard"|"soft",
"highestTierAllowed": t}) is confined to ONE network domain (hypernode)
of tier ≤ t.  MI355X mapping: every hypernode's member set is a dynamic
bit plane; the per-job domain choice (the "gradient" — prefer the
lowest-tier feasible domain, then the one already hosting the job's
tasks) is a cheap host-side scan over per-domain free-capacity vectors,
and the chosen domain rides the predicate kernel as one require bit.

Hard mode: no feasible domain ⇒ the job's classes get an impossible
require bit (never placed this cycle).  Soft mode: feasible domain
preferred, otherwise unconstrained.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np

from ...api.hypernode import HyperNodeTree
from ..tensors import set_plane_bit
from .base import Plugin, register


@register("network-topology-aware")
class NetworkTopologyAwarePlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        store = getattr(ssn.cache, "store", None)
        hypernodes = store.list("HyperNode") if store is not None else []
        if not hypernodes:
            return
        nt = ssn.node_tensors
        tree = HyperNodeTree(
            hypernodes, list(ssn.nodes),
            {n: ni.node.meta.labels for n, ni in ssn.nodes.items()})
        ssn.hypernode_tree = tree

        # per-domain: node ids, free capacity vector, membership bit
        self.domain_ids: Dict[str, list] = {}
        self.domain_bit: Dict[str, int] = {}
        free = {}
        for hname, nodes in tree.members.items():
            ids = [ssn.nodes[n].node_id for n in nodes if n in ssn.nodes]
            self.domain_ids[hname] = ids
            self.domain_bit[hname] = nt.add_dynamic_bit(f"hn:{hname}", ids)
            vec = np.zeros(nt.r, dtype=np.float64)
            for n in nodes:
                ni = ssn.nodes.get(n)
                if ni is not None:
                    vec += nt.resource_vector(ni.idle)
            free[hname] = vec

        impossible_bit = nt.add_dynamic_bit("hn:none", [])
        choice: Dict[str, Optional[str]] = {}

        def choose_domain(job) -> Optional[str]:
            if job.key in choice:
                return choice[job.key]
            spec = job.podgroup.spec.network_topology if job.podgroup else None
            if not spec:
                choice[job.key] = None
                return None
            max_tier = spec.get("highestTierAllowed")
            need = np.zeros(nt.r, dtype=np.float64)
            for t in job.pending_tasks:
                v = nt.req_vector(t)
                if v is not None:
                    need += v
            # nodes already hosting the job pin candidate domains
            running_nodes = {t.node_name for t in job.tasks.values()
                             if t.node_name}
            best = None
            for hname in tree.domains_by_tier(max_tier):
                members = tree.members[hname]
                if running_nodes and not running_nodes <= members:
                    continue
                if (free[hname] + 0.1 >= need).all():
                    best = hname
                    break
            choice[job.key] = best
            return best

        # per-SUBGROUP domains (SubGroupPolicy networkTopology): classes
        # carrying a topology override pick their own domain against a
        # build-time reservation ledger, so several subgroups of one job
        # spread across racks within ONE cycle (a failed subgroup's
        # reservation is simply re-chosen next cycle)
        free_resv = {h: f.copy() for h, f in free.items()}

        def choose_domain_for_class(tclass):
            spec = tclass.topology
            max_tier = spec.get("highestTierAllowed")
            v = nt.req_vector(tclass.tasks[0])
            if v is None:
                return None
            need = v.astype(np.float64) * len(tclass.tasks)
            for hname in tree.domains_by_tier(max_tier):
                if (free_resv[hname] + 0.1 >= need).all():
                    free_resv[hname] = free_resv[hname] - need
                    return hname
            return None

        def hook(tclass, job, require, forbid):
            spec2 = getattr(tclass, "topology", None)
            if spec2:
                best = choose_domain_for_class(tclass)
                if best is not None:
                    set_plane_bit(require, self.domain_bit[best])
                elif spec2.get("mode", "hard") == "hard":
                    set_plane_bit(require, impossible_bit)
                return
            if job is None or job.podgroup is None:
                return
            spec = job.podgroup.spec.network_topology
            if not spec:
                return
            best = choose_domain(job)
            if best is not None:
                set_plane_bit(require, self.domain_bit[best])
            elif spec.get("mode", "hard") == "hard":
                set_plane_bit(require, impossible_bit)
            # soft + no feasible domain: unconstrained

        hook.is_topology = True       # preempt domain trials strip it
        domain_sets = {h: set(ids) for h, ids in self.domain_ids.items()}

        def on_allocate(tclass, node_ids, counts, tasks=None):
            """Keep per-domain free capacity current within the cycle."""
            if not tclass.tasks:
                return
            req = nt.req_vector(tclass.tasks[0])
            if req is None:
                return
            placed_per_node = dict(zip(node_ids, counts))
            for hname, ids in domain_sets.items():
                hit = sum(c for nid, c in placed_per_node.items()
                          if nid in ids)
                if hit:
                    free[hname] -= hit * req.astype(np.float64)

        handler = type("NtaHandler", (), {"on_allocate":
                                          staticmethod(on_allocate)})()
        ssn.class_constraint_hooks.append(hook)
        ssn.event_handlers.append(handler)
        self.tree = tree
