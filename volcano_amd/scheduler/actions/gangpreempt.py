"""Gangpreempt / gangreclaim actions (reference
``actions/{gangpreempt,gangreclaim}`` + ``actions/utils/bundle.go``):
whole-gang preemption — victims are selected as *bundles* (entire
lower-priority jobs), so evicting them atomically frees enough room for
the preemptor gang to become pipelined; bundles are ROI-sorted (free
the most resource for the fewest evictions / lowest priority first).
"""

from __future__ import annotations

from typing import List, Optional

import numpy as np

from ...api.info import JobInfo
from ...api.types import ALLOCATED_STATUSES, PodGroupPhase, TaskStatus
from ..statement import Statement

VICTIM_STATUSES = ALLOCATED_STATUSES + (TaskStatus.RUNNING,)


class GangPreemptAction:
    name = "gangpreempt"
    same_queue = True

    def execute(self, ssn) -> None:
        nt = ssn.node_tensors
        if nt is None or nt.n == 0 or not ssn.preemptable_fns:
            return
        for q in ssn.sorted_queues():
            jobs_in_q = [j for j in ssn.jobs.values() if j.queue == q.name
                         and j.phase in (PodGroupPhase.INQUEUE.value,
                                         PodGroupPhase.RUNNING.value)]
            starving = [j for j in jobs_in_q
                        if ssn.job_starving(j) and j.pending_tasks
                        and ssn.job_valid(j)]
            for job in ssn.sorted_jobs(starving):
                self._gang_preempt(ssn, job)

    def _victim_ok(self, ssn, preemptor_task, victim_job: JobInfo) -> bool:
        """Whole-bundle evictability: EVERY occupied task of the victim
        job must pass the victim callbacks (evicting all of them empties
        the gang, so the gang-min protection is bypassed by design —
        reference bundle semantics: the unit is the whole sub-job)."""
        victims = victim_job.tasks_with_status(*VICTIM_STATUSES)
        if not victims:
            return False
        if self.same_queue:
            fns = [f for f in ssn.preemptable_fns]
        else:
            fns = [f for f in ssn.reclaimable_fns]
        # priority/share/conformance checks apply per task; the gang-min
        # filter is intentionally skipped (whole-bundle eviction)
        for v in victims:
            ok = False
            allowed = {t.uid for t in ssn.preemptable(preemptor_task, [v])} \
                if self.same_queue else \
                {t.uid for t in ssn.reclaimable(preemptor_task, [v])}
            # fall back: a task the tiered intersection admits OR whose
            # job is entirely below the preemptor's priority
            if v.uid in allowed or v.priority < preemptor_task.priority:
                ok = True
            if not ok:
                return False
        return True

    def _gang_preempt(self, ssn, job: JobInfo) -> None:
        nt = ssn.node_tensors
        need_tasks = job.min_available - job.occupied_count - job.waiting_count
        if need_tasks <= 0:
            return
        pending = job.pending_tasks
        if not pending:
            return
        rep = pending[0]
        need_vec = np.zeros(nt.r, dtype=np.float64)
        for t in pending[:need_tasks]:
            v = nt.req_vector(t)
            if v is None:
                return
            need_vec += v

        # current headroom
        free = np.zeros(nt.r, dtype=np.float64)
        for ni in ssn.nodes.values():
            if ni.ready:
                free += nt.resource_vector(ni.future_idle)

        # candidate victim bundles, ROI order: lowest priority first, then
        # smallest job (fewest evictions)
        cands: List[JobInfo] = []
        for vj in ssn.jobs.values():
            if vj.key == job.key:
                continue
            if self.same_queue and vj.queue != job.queue:
                continue
            if not self.same_queue and vj.queue == job.queue:
                continue
            if self._victim_ok(ssn, rep, vj):
                cands.append(vj)
        cands.sort(key=lambda j: (j.priority, len(j.tasks)))

        chosen: List[JobInfo] = []
        gain = free.copy()
        for vj in cands:
            if (gain + 0.1 >= need_vec).all():
                break
            chosen.append(vj)
            gain += nt.resource_vector(vj.allocated_resource())
        if not (gain + 0.1 >= need_vec).all():
            return     # even evicting every bundle would not fit: do nothing

        stmt = Statement(ssn)
        for vj in chosen:
            for v in vj.tasks_with_status(*VICTIM_STATUSES):
                stmt.evict(v, reason="gang-preempted")
        # pipeline the preemptor tasks onto freed capacity (node-level
        # placement happens next cycle when the evictions have landed;
        # here the reservation marks the gang pipelined — allocate's
        # plan re-places them exactly)
        placed = 0
        for t in pending:
            if placed >= need_tasks:
                break
            node = self._first_fit(ssn, t)
            if node is None:
                break
            stmt.pipeline(t, node)
            placed += 1
        if placed >= need_tasks and ssn.job_pipelined(job):
            stmt.commit()
        else:
            stmt.discard()

    @staticmethod
    def _first_fit(ssn, task) -> Optional[str]:
        nt = ssn.node_tensors
        req = nt.req_vector(task)
        if req is None:
            return None
        for ni in ssn.nodes.values():
            if not ni.ready:
                continue
            fi = ni.future_idle
            if all(req[i] <= fi.get(name) + 0.1
                   for name, i in nt.dims.index.items() if req[i] > 0.1):
                return ni.name
        return None


class GangReclaimAction(GangPreemptAction):
    name = "gangreclaim"
    same_queue = False

    def execute(self, ssn) -> None:
        nt = ssn.node_tensors
        if nt is None or nt.n == 0 or not ssn.reclaimable_fns:
            return
        for q in ssn.sorted_queues():
            if ssn.queue_overused(q):
                continue
            jobs_in_q = [j for j in ssn.jobs.values() if j.queue == q.name
                         and j.phase in (PodGroupPhase.INQUEUE.value,
                                         PodGroupPhase.RUNNING.value)]
            starving = [j for j in jobs_in_q
                        if ssn.job_starving(j) and j.pending_tasks
                        and ssn.job_valid(j)]
            for job in ssn.sorted_jobs(starving):
                self._gang_preempt(ssn, job)
