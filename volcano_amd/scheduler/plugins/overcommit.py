"""Overcommit plugin (reference ``plugins/overcommit``): admit jobs into
the Inqueue state while total inqueue resource <= cluster total * factor
(default 1.2)."""

from __future__ import annotations

import torch

from ..session import ABSTAIN, PERMIT, REJECT
from .base import Plugin, register


@register("overcommit")
class OvercommitPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        factor = float(self.args.get("overcommit-factor", 1.2))
        nt = ssn.node_tensors
        total = (ssn.total_resource if ssn.total_resource is not None
                 else torch.zeros(nt.r)) * factor
        # resource already admitted: inqueue + running job requests
        from ...api.types import PodGroupPhase
        inqueue = torch.zeros(nt.r, dtype=torch.float32)
        for job in ssn.jobs.values():
            if job.phase in (PodGroupPhase.INQUEUE.value,
                             PodGroupPhase.RUNNING.value):
                inqueue += torch.from_numpy(
                    nt.resource_vector(job.total_request()))
        state = {"inqueue": inqueue}

        def job_enqueueable(job) -> int:
            minres = torch.from_numpy(nt.resource_vector(
                job.podgroup.spec.min_resources)) if job.podgroup else \
                torch.zeros(nt.r)
            mask = total > 0
            if bool(((state["inqueue"] + minres)[mask] <= total[mask] + 0.1).all()):
                return PERMIT
            return REJECT

        def job_enqueued(job) -> None:
            minres = torch.from_numpy(nt.resource_vector(
                job.podgroup.spec.min_resources)) if job.podgroup else \
                torch.zeros(nt.r)
            state["inqueue"] += minres

        ssn.job_enqueueable_fns.append(job_enqueueable)
        ssn.job_enqueued_fns = getattr(ssn, "job_enqueued_fns", [])
        ssn.job_enqueued_fns.append(job_enqueued)
