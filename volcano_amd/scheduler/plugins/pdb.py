"""PDB plugin (reference ``plugins/pdb``): honors PodDisruptionBudgets —
filters victims whose eviction would push a budget's matched, healthy pod
count below minAvailable (or disruptions above maxUnavailable)."""

from __future__ import annotations

from typing import List

from ...api.types import TaskStatus
from .base import Plugin, register


@register("pdb")
class PdbPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        from ...utils.features import enabled
        if not enabled("PodDisruptionBudgetsSupport"):
            return
        store = getattr(ssn.cache, "store", None)
        pdbs = store.list("PodDisruptionBudget") if store is not None else []
        if not pdbs:
            return

        def matched(pdb, task) -> bool:
            pod = task.pod
            if pod is None:
                return False
            return all(pod.meta.labels.get(k) == v
                       for k, v in pdb.selector.items())

        def budget_filter(evictor, candidates: List) -> List:
            out = []
            taken: dict = {}
            for v in candidates:
                ok = True
                for pdb in pdbs:
                    if not matched(pdb, v):
                        continue
                    healthy = sum(
                        1 for job in ssn.jobs.values()
                        for t in job.tasks.values()
                        if t.status in (TaskStatus.RUNNING, TaskStatus.BOUND)
                        and matched(pdb, t))
                    healthy -= taken.get(id(pdb), 0)
                    if pdb.min_available is not None and \
                            healthy - 1 < pdb.min_available:
                        ok = False
                        break
                    if pdb.max_unavailable is not None and \
                            taken.get(id(pdb), 0) + 1 > pdb.max_unavailable:
                        ok = False
                        break
                if ok:
                    out.append(v)
                    for pdb in pdbs:
                        if matched(pdb, v):
                            taken[id(pdb)] = taken.get(id(pdb), 0) + 1
            return out

        ssn.preemptable_fns.append(budget_filter)
        ssn.reclaimable_fns.append(budget_filter)
        ssn.victim_filter_fns.append(lambda victims: budget_filter(None, victims))
