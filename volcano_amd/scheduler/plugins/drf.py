"""DRF plugin (reference ``plugins/drf/drf.go``).

Dominant share per job — ``share = max_r(allocated_r / total_r)``
(drf.go calculateShare) — computed as ONE tensor pass over the [J, R]
allocation matrix (ops.reference.drf_share / float64), not per-job Go
loops.  JobOrder: lower share first (:388).  Preemptable: victim's job
share must stay above the preemptor's (:263).
"""

from __future__ import annotations

import torch

from ...ops import reference as ref
from .base import Plugin, register


@register("drf")
class DrfPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        jobs = list(ssn.jobs.values())
        J, R = len(jobs), nt.r
        alloc = torch.zeros((J, R), dtype=torch.float32)
        for i, job in enumerate(jobs):
            alloc[i] = torch.from_numpy(nt.resource_vector(job.allocated_resource()))
        total = ssn.total_resource if ssn.total_resource is not None else \
            torch.ones(R)
        shares = ref.drf_share(alloc, total)          # [J] f64
        self.share = {job.key: float(shares[i]) for i, job in enumerate(jobs)}
        self.alloc = {job.key: alloc[i] for i, job in enumerate(jobs)}
        self.total = total

        def job_order(a, b) -> int:
            sa = self.share.get(a.key, 0.0)
            sb = self.share.get(b.key, 0.0)
            if abs(sa - sb) < 1e-12:
                return 0
            return -1 if sa < sb else 1

        def preemptable(preemptor, candidates):
            ps = self.share.get(preemptor.job_key, 0.0)
            out = []
            for v in candidates:
                vs = self.share.get(v.job_key, 0.0)
                if vs > ps:      # victim's job is richer than preemptor's
                    out.append(v)
            return out

        class Handler:
            """Keep shares current as the cycle stages allocations."""
            def on_allocate(h, tclass, node_ids, counts):
                key = tclass.tasks[0].job_key if tclass.tasks else None
                if key is None or key not in self.alloc:
                    return
                n = sum(counts)
                vec = torch.from_numpy(nt.resource_vector(tclass.request))
                self.alloc[key] = self.alloc[key] + n * vec
                t = torch.clamp(self.total.to(torch.float64), min=1.0)
                self.share[key] = float(
                    (self.alloc[key].to(torch.float64) / t).amax())

        ssn.add_job_order_fn(job_order)
        ssn.preemptable_fns.append(preemptable)
        ssn.event_handlers.append(Handler())
