"""Task-topology plugin (reference ``plugins/task-topology``, design
``docs/design/task-topology-plugin.md``): task affinity/anti-affinity
*within* a job — e.g. ps/worker pairs packed together, or replicas of a
role spread apart.

Arguments (job annotation ``volcano.sh/task-topology``-style, passed as
plugin args here): {"affinity": [["ps","worker"]],
                    "anti-affinity": [["worker"]]}

MI355X mapping: affinity = per-job *score bias* toward nodes already
hosting the job's tasks (the bias plane is rebuilt per job group between
kernel passes via the allocate event stream); anti-affinity = the class
spreads by preferring empty nodes (the kernel's least-requested term
already does that) plus a TaskOrder that schedules affine roles
back-to-back so their classes see each other's placements.
"""

from __future__ import annotations

from typing import Dict, List, Set

from .base import Plugin, register


@register("task-topology")
class TaskTopologyPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        affinity: List[List[str]] = self.args.get("affinity", []) or []
        anti: List[List[str]] = self.args.get("anti-affinity", []) or []
        pairs: Set[tuple] = set()
        for group in affinity:
            for a in group:
                for b in group:
                    pairs.add((a, b))
        self.affine_roles = pairs
        anti_roles = {r for group in anti for r in group}

        # affine roles schedule adjacently (bucket order); reference
        # topology.go:345 TaskOrder
        rank: Dict[str, int] = {}
        for i, group in enumerate(affinity):
            for r in group:
                rank.setdefault(r, i)

        def task_order(a, b) -> int:
            ra = rank.get(a.role, len(affinity))
            rb = rank.get(b.role, len(affinity))
            if ra != rb:
                return -1 if ra < rb else 1
            return 0

        ssn.add_task_order_fn(task_order)

        # anti-affinity roles: spread via the least-requested direction
        if anti_roles:
            ssn.score_weights["least"] = max(
                ssn.score_weights.get("least", 0.0), 1.0)

        # -- bucket scoring (reference topology.go:138 calcBucketScore +
        # :193 NodeOrderFn): classes of affine roles get a per-class score
        # bias toward nodes already hosting the job's bucket members, and
        # away from nodes hosting anti-affine peers.  The bias plane rides
        # the kernel's additive bias input (ClassPlan.bias), so bucket
        # preference is evaluated for ALL nodes in the same fused pass as
        # feasibility — no host (task, node) callbacks.
        import numpy as np
        from ...api.types import TaskStatus
        weight = float(self.args.get("task-topology.weight", 10.0))
        nt = ssn.node_tensors
        group_memo: Dict[str, tuple] = {}

        def job_groups(job):
            """(affinity groups, anti groups) for a job: plugin args plus
            the reference's podgroup annotations
            (util.go:36 volcano.sh/task-topology-affinity)."""
            got = group_memo.get(job.key)
            if got is not None:
                return got
            aff_g = [list(g) for g in affinity]
            anti_g = [list(g) for g in anti]
            ann = job.podgroup.meta.annotations if job.podgroup else {}
            a = ann.get("volcano.sh/task-topology-affinity")
            if a:
                aff_g += [s.split(",") for s in a.split(";") if s]
            b = ann.get("volcano.sh/task-topology-anti-affinity")
            if b:
                anti_g += [s.split(",") for s in b.split(";") if s]
            got = group_memo[job.key] = (aff_g, anti_g)
            return got

        def class_bias(tclass, job):
            if job is None or nt is None or nt.n == 0:
                return None
            aff_g, anti_g = job_groups(job)
            if not aff_g and not anti_g:
                return None
            role = tclass.role
            partners = set()
            for g in aff_g:
                if role in g:
                    partners.update(g)
            repel = set()
            for g in anti_g:
                if role in g:
                    repel.update(g)
            if not partners and not repel:
                return None
            bias = None
            for t in job.tasks.values():
                if not t.node_name:
                    continue
                if not (t.status.occupies_node
                        or t.status == TaskStatus.PIPELINED):
                    continue
                w = (weight if t.role in partners else 0.0) \
                    - (weight if t.role in repel else 0.0)
                if w == 0.0:
                    continue
                ni = ssn.nodes.get(t.node_name)
                if ni is None or ni.node_id < 0:
                    continue
                if bias is None:
                    bias = np.zeros(nt.n, dtype=np.float32)
                bias[ni.node_id] += w
            return bias

        ssn.class_bias_fns.append(class_bias)
