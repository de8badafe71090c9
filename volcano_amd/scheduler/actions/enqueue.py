"""Enqueue action (reference ``actions/enqueue/enqueue.go:44-105``).

Admission control: pending PodGroups become Inqueue when the enqueueable
vote (proportion/overcommit/sla) passes — queues popped in queue order,
jobs in job order.
"""

from __future__ import annotations

from ...api.types import PodGroupPhase


class EnqueueAction:
    name = "enqueue"

    def execute(self, ssn) -> None:
        by_queue = {}
        for job in ssn.jobs.values():
            if job.phase != PodGroupPhase.PENDING.value:
                continue
            by_queue.setdefault(job.queue, []).append(job)
        queues = [ssn.queues[q] for q in by_queue if q in ssn.queues]
        for q in ssn.sorted_queues(queues):
            if not q.is_open:
                continue
            for job in ssn.sorted_jobs(by_queue[q.name]):
                if ssn.job_enqueueable(job):
                    if job.podgroup is not None:
                        job.podgroup.status.phase = PodGroupPhase.INQUEUE.value
                    for fn in getattr(ssn, "job_enqueued_fns", []):
                        fn(job)
                    ssn.cache.update_podgroup(job)
