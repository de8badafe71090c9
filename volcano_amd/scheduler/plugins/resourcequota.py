"""Resourcequota plugin (reference ``plugins/resourcequota``): rejects
job enqueue when the namespace's ResourceQuota cannot hold the job's
min resources on top of what the namespace already uses."""

from __future__ import annotations

from ...api.resource import Resource
from ...api.types import PodGroupPhase
from ..session import ABSTAIN, REJECT
from .base import Plugin, register


@register("resourcequota")
class ResourceQuotaPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        store = getattr(ssn.cache, "store", None)
        quotas = store.list("ResourceQuota") if store is not None else []
        if not quotas:
            return
        by_ns = {}
        for q in quotas:
            by_ns.setdefault(q.meta.namespace, []).append(q)

        # namespace usage: requests of pods that hold or will hold resources
        used = {}
        for job in ssn.jobs.values():
            ns = job.namespace
            r = used.setdefault(ns, Resource())
            r.add(job.allocated_resource())
            if job.phase == PodGroupPhase.INQUEUE.value:
                if job.podgroup is not None:
                    r.add(job.podgroup.spec.min_resources)

        def job_enqueueable(job) -> int:
            ns_quotas = by_ns.get(job.namespace)
            if not ns_quotas:
                return ABSTAIN
            minres = job.podgroup.spec.min_resources if job.podgroup \
                else Resource()
            ns_used = used.get(job.namespace, Resource())
            for q in ns_quotas:
                trial = ns_used.clone().add(minres)
                if not trial.less_equal(q.hard):
                    return REJECT
            return ABSTAIN

        def job_enqueued(job) -> None:
            if job.podgroup is not None:
                used.setdefault(job.namespace, Resource()).add(
                    job.podgroup.spec.min_resources)

        ssn.job_enqueueable_fns.insert(0, job_enqueueable)
        ssn.job_enqueued_fns = getattr(ssn, "job_enqueued_fns", [])
        ssn.job_enqueued_fns.append(job_enqueued)
