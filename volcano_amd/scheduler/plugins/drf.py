"""DRF plugin (reference ``plugins/drf/drf.go``).

Dominant share per job — ``share = max_r(allocated_r / total_r)``
(drf.go calculateShare) — computed as ONE vectorized pass over the [J, R]
allocation matrix (float64), not per-job Go loops.  JobOrder: lower share
first (:388).  Preemptable: victim's job share must stay above the
preemptor's (:263).

Note on the reference's share-update event handlers (drf.go:391): the
reference mutates shares *during* the cycle because its allocation is
incremental per task.  In the plan design the job order is frozen when
the cycle plan is built, so mid-cycle share updates cannot influence any
decision — shares are recomputed at the next session open instead.
"""

from __future__ import annotations

import numpy as np

from .base import Plugin, register


@register("drf")
class DrfPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        jobs = list(ssn.jobs.values())
        if not jobs:
            return
        # zero-alloc jobs (the common start-of-cycle state) skip the
        # vector build entirely — share is exactly 0
        alloc = np.zeros((len(jobs), nt.r), dtype=np.float64)
        for i, job in enumerate(jobs):
            if job.occupied_count:
                alloc[i] = job.alloc_vec(nt)
        total = ssn.total_resource.numpy().astype(np.float64) \
            if ssn.total_resource is not None else np.ones(nt.r)
        shares = (alloc / np.maximum(total, 1.0)).max(axis=1)     # [J]
        self.share = {job.key: float(shares[i]) for i, job in enumerate(jobs)}

        def job_order(a, b) -> int:
            sa = self.share.get(a.key, 0.0)
            sb = self.share.get(b.key, 0.0)
            if abs(sa - sb) < 1e-12:
                return 0
            return -1 if sa < sb else 1

        def preemptable(preemptor, candidates):
            ps = self.share.get(preemptor.job_key, 0.0)
            return [v for v in candidates
                    if self.share.get(v.job_key, 0.0) > ps]

        ssn.add_job_order_fn(
            job_order, key=lambda j: self.share.get(j.key, 0.0))
        ssn.preemptable_fns.append(preemptable)
