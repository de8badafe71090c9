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
        jt = getattr(ssn, "job_table", None)
        if jt is not None and len(jt.jobs) == len(ssn.jobs):
            jobs = jt.jobs
            occ_rows = np.nonzero(jt.occ > 0)[0]
        else:
            jobs = list(ssn.jobs.values())
            occ_rows = np.array([i for i, j in enumerate(jobs)
                                 if j.occupied_count], dtype=np.int64)
        if not len(jobs):
            return
        # zero-alloc jobs (the common start-of-cycle state) skip the
        # vector build entirely — share is exactly 0
        if not len(occ_rows):
            self.share = {}
        else:
            alloc = np.zeros((len(jobs), nt.r), dtype=np.float64)
            for i in occ_rows:
                alloc[i] = jobs[i].alloc_vec(nt)
            total = ssn.total_resource.numpy().astype(np.float64) \
                if ssn.total_resource is not None else np.ones(nt.r)
            shares = (alloc / np.maximum(total, 1.0)).max(axis=1)     # [J]
            self.share = {job.key: float(shares[i])
                          for i, job in enumerate(jobs)}

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

        any_share = bool(self.share) and any(
            v != 0.0 for v in self.share.values())

        def share_col(jt, rows):
            if not any_share:
                return np.zeros(len(rows))
            sh = self.share
            return np.fromiter((sh.get(k, 0.0) for k in jt.keys[rows]),
                               dtype=np.float64, count=len(rows))

        ssn.add_job_order_fn(
            job_order, key=lambda j: self.share.get(j.key, 0.0),
            col=share_col)
        ssn.preemptable_fns.append(preemptable)
