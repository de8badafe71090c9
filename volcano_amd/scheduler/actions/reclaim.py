"""Reclaim action (reference ``actions/reclaim/reclaim.go:57-264``).

Cross-queue fairness enforcement: a starving job in a queue under its
deserved share reclaims resources from queues over their share — victims
come from the Reclaimable callbacks (proportion: only overused queues and
only if the victim's queue is ``reclaimable``; gang protects minAvailable;
conformance protects critical pods).  Shares the Statement machinery with
preempt (same skeleton, different victim scope — SURVEY §3.4).
"""

from __future__ import annotations

from ...api.types import PodGroupPhase
from .preempt import PreemptAction


class ReclaimAction(PreemptAction):
    name = "reclaim"

    def execute(self, ssn) -> None:
        nt = ssn.node_tensors
        if nt is None or nt.n == 0 or not ssn.reclaimable_fns:
            return
        for q in ssn.sorted_queues():
            if not q.is_open:
                continue        # closed queues cannot reclaim
                # (reference reclaim.go queue state check)
            if ssn.queue_overused(q):
                continue        # only under-served queues reclaim
            jobs_in_q = [j for j in ssn.jobs.values() if j.queue == q.name
                         and j.phase in (PodGroupPhase.INQUEUE.value,
                                         PodGroupPhase.RUNNING.value)]
            starving = [j for j in jobs_in_q
                        if ssn.job_starving(j) and j.pending_tasks
                        and ssn.job_valid(j)]
            for job in ssn.sorted_jobs(starving):
                self._preempt_for_job(
                    ssn, job, same_queue=False,
                    victim_filter=lambda t, cands: ssn.reclaimable(t, cands))
