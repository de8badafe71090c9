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
