"""Overcommit plugin (reference ``plugins/overcommit``): admit jobs into
the Inqueue state while total inqueue resource <= cluster total * factor
(default 1.2).  Numpy scalar ops per vote (hot at 10k+ jobs/cycle)."""

from __future__ import annotations

import numpy as np

from ...api.types import PodGroupPhase
from ..session import PERMIT, REJECT
from .base import Plugin, register


@register("overcommit")
class OvercommitPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        factor = float(self.args.get("overcommit-factor", 1.2))
        nt = ssn.node_tensors
        total = (ssn.total_resource.numpy() if ssn.total_resource is not None
                 else np.zeros(nt.r, dtype=np.float32)) * factor
        total = total.astype(np.float64)
        mask = total > 0
        # admitted-but-not-yet-running work (reference overcommit counts
        # Inqueue-phase podgroups; Running jobs hold real allocations and
        # no longer occupy admission headroom)
        inqueue = np.zeros(nt.r, dtype=np.float64)
        jt = getattr(ssn, "job_table", None)
        if jt is not None and len(jt.jobs) == len(ssn.jobs) \
                and jt.minres.shape[1] == nt.r:
            from ..jobtable import PH_INQUEUE
            sel = jt.phase == PH_INQUEUE
            if sel.any():
                inqueue += jt.minres[sel].sum(axis=0)
        else:
            for job in ssn.jobs.values():
                if job.phase == PodGroupPhase.INQUEUE.value:
                    inqueue += job.minres_vec(nt)

        def job_enqueueable(job) -> int:
            minres = job.minres_vec(nt)
            if (minres[~mask] > 0.1).any():
                return REJECT     # demand on a zero-capacity dim
            head = inqueue[mask] + minres[mask]
            ok = bool((head <= total[mask] + 0.1 + 1e-6 * total[mask]).all())
            return PERMIT if ok else REJECT

        def job_enqueued(job) -> None:
            inqueue[:] += job.minres_vec(nt)

        def job_enqueueable_bulk(qname, jobs, rows=None, table=None):
            """Whole-batch admission: total demand fits <=> every prefix
            fits (monotone sum), so admitting all == per-job votes.
            Returns a commit thunk (run only when all plugins agree)."""
            if rows is not None and table is not None \
                    and table.minres.shape[1] == nt.r:
                demand = table.minres[rows].sum(axis=0)
            else:
                demand = np.zeros(nt.r, dtype=np.float64)
                for j in jobs:
                    demand += j.minres_vec(nt)
            if (demand[~mask] > 0.1).any():
                return None       # demand on a zero-capacity dim
            head = inqueue[mask] + demand[mask]
            if bool((head <= total[mask] + 0.1 + 1e-6 * total[mask]).all()):
                return lambda: inqueue.__iadd__(demand)
            return None

        ssn.job_enqueueable_fns.append(job_enqueueable)
        ssn.job_enqueueable_bulk_fns.append(job_enqueueable_bulk)
        ssn.job_enqueued_fns = getattr(ssn, "job_enqueued_fns", [])
        ssn.job_enqueued_fns.append(job_enqueued)
