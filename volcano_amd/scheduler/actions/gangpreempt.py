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

    def _victim_ok(self, ssn, preemptor_task, preemptor_job: JobInfo,
                   victim_job: JobInfo) -> bool:
        """Whole-bundle evictability (reference gangpreempt.go:189-246):
        EVERY occupied task of the victim job must pass the victim
        plugin intersection — the plugin vetoes (PDB budget, conformance
        kube-system protection, cooldown) are authoritative, with ONE
        exception: the gang plugin's keep-above-minAvailable filter is
        bypassed, because the unit of eviction is the whole bundle
        (evicting all of a gang's tasks never strands it below min).
        Additionally the reference requires (a) the victim job strictly
        below the preemptor's priority, (b) the victim job not explicitly
        marked non-preemptable (unset defaults to allowed —
        utils.IsJobPreemptableForGangEviction, actions/utils/util.go:114),
        and (c) no victim pod explicitly annotated non-preemptable
        (GetPodPreemptable defaults true, api/pod_info.go:176)."""
        victims = victim_job.tasks_with_status(*VICTIM_STATUSES)
        if not victims:
            return False
        if victim_job.priority >= preemptor_job.priority:
            return False
        pg = victim_job.podgroup
        if pg is not None:
            from ...api.objects import ANN_PREEMPTABLE
            ann = pg.meta.annotations.get(ANN_PREEMPTABLE) or \
                pg.meta.labels.get(ANN_PREEMPTABLE)
            if ann is not None and ann != "true":
                return False
        for v in victims:
            if not v.preemptable:   # explicit volcano.sh/preemptable=false
                return False
        fns = ssn.preemptable_fns if self.same_queue else ssn.reclaimable_fns
        allowed = victims
        for fn in fns:
            if getattr(fn, "bundle_exempt", False):
                continue    # gang-min protection — whole-bundle semantics
            keep = {t.uid for t in fn(preemptor_task, allowed)}
            allowed = [t for t in allowed if t.uid in keep]
            if len(allowed) < len(victims):
                return False    # some task vetoed → the bundle is not free
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
            if self._victim_ok(ssn, rep, job, vj):
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
