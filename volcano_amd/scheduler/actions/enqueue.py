"""Enqueue action (reference ``actions/enqueue/enqueue.go:44-105``).

Admission control: pending PodGroups become Inqueue when the enqueueable
vote (proportion/overcommit/sla) passes — queues popped in queue order,
jobs in job order.  Also plays the gate manager (reference
``gate/schedulinggate.go:43-113`` + feature gate
SchedulingGatesQueueAdmission): the ``volcano.sh/queue-allocation-gate``
scheduling gate is lifted from a job's pods once its queue admits it.
"""

from __future__ import annotations

from ...api.types import PodGroupPhase

QUEUE_GATE = "volcano.sh/queue-allocation-gate"


class EnqueueAction:
    name = "enqueue"

    def execute(self, ssn) -> None:
        jt = getattr(ssn, "job_table", None)
        if jt is not None and len(jt.jobs) == len(ssn.jobs):
            self._execute_columnar(ssn, jt)
            return
        by_queue = {}
        for job in ssn.jobs.values():
            if job.phase != PodGroupPhase.PENDING.value:
                continue
            by_queue.setdefault(job.queue, []).append(job)
        queues = [ssn.queues[q] for q in by_queue if q in ssn.queues]
        for q in ssn.sorted_queues(queues):
            if not q.is_open:
                continue
            jobs_q = by_queue[q.name]
            self._admit_queue(ssn, q, jobs_q, rows=None, jt=None)

    def _execute_columnar(self, ssn, jt) -> None:
        """Selection + grouping over the JobTable columns; admission per
        queue in queue order (same semantics as the per-job path)."""
        import numpy as np
        from ..jobtable import PH_PENDING
        sel = np.nonzero(jt.phase == PH_PENDING)[0]
        if len(sel) and ssn.job_enqueueable_fns:
            # minResources naming a resource no dim represents can never
            # fit realCapability (reference proportion jobEnqueueableFn)
            sel = sel[~jt.minres_bad[sel]]
        if not len(sel):
            return
        qis = jt.qi[sel]
        qname_by_qi = {i: n for n, i in ssn.queue_index.items()}
        present = np.unique(qis[qis >= 0])
        queues = [ssn.queues[qname_by_qi[int(i)]] for i in present
                  if qname_by_qi.get(int(i)) in ssn.queues]
        for q in ssn.sorted_queues(queues):
            if not q.is_open:
                continue
            qi = ssn.queue_index[q.name]
            rows = sel[qis == qi]
            self._admit_queue(ssn, q, None, rows=rows, jt=jt)

    def _admit_queue(self, ssn, q, jobs_q, rows=None, jt=None) -> None:
        if jobs_q is None:
            jobs_q = [jt.jobs[int(k)] for k in rows]
        elif ssn.job_enqueueable_fns and ssn.node_tensors is not None:
            # non-columnar path: same unrepresented-resource admission
            # veto the columnar path applies via jt.minres_bad
            jobs_q = [j for j in jobs_q
                      if not j.minres_unrepresented(ssn.node_tensors)]
        if not jobs_q:
            return
        fifo = q.queue.spec.dequeue_strategy == "fifo"
        # bulk fast path: when every enqueueable plugin can admit the
        # whole batch (total demand fits), skip the per-job votes —
        # identical outcome for monotone-sum admission
        bulk = ssn.job_enqueueable_bulk_fns
        store = getattr(ssn.cache, "store", None)
        if bulk and len(bulk) == len(ssn.job_enqueueable_fns):
            # each fn returns a commit thunk iff the whole batch fits;
            # commit only once EVERY plugin agreed (no partial updates)
            commits = [fn(q.name, jobs_q, rows=rows, table=jt)
                       for fn in bulk]
            if all(c is not None for c in commits):
                for c in commits:
                    c()
                inq = PodGroupPhase.INQUEUE.value
                from ..jobtable import PH_INQUEUE
                for job in jobs_q:
                    if job.podgroup is not None:
                        job.podgroup.status.phase = inq
                    if jt is not None:
                        jt.phase[job._jrow] = PH_INQUEUE
                    if store is not None:
                        ssn.cache.update_podgroup(job)
                    if getattr(job, "_gated", 0):
                        self._lift_queue_gates(ssn, job)
                return
        if rows is not None:
            rows = ssn.ordered_job_rows(jt, rows)
            ordered = (jt.jobs[int(k)] for k in rows)
        else:
            ordered = ssn.sorted_jobs(jobs_q)
        from ..jobtable import PH_INQUEUE
        for job in ordered:
            if ssn.job_enqueueable(job):
                if job.podgroup is not None:
                    job.podgroup.status.phase = PodGroupPhase.INQUEUE.value
                if jt is not None:
                    jt.phase[job._jrow] = PH_INQUEUE
                for fn in getattr(ssn, "job_enqueued_fns", []):
                    fn(job)
                ssn.cache.update_podgroup(job)
                self._lift_queue_gates(ssn, job)
            elif fifo:
                break   # head-of-line blocks the queue (dequeueStrategy)

    @staticmethod
    def _lift_queue_gates(ssn, job) -> None:
        if not job.has_gated_tasks:
            return
        from ...utils.features import enabled
        if not enabled("SchedulingGatesQueueAdmission"):
            return
        store = getattr(ssn.cache, "store", None)
        for t in job.tasks.values():
            if not t.gated or t.pod is None:
                continue
            gates = t.pod.scheduling_gates
            if QUEUE_GATE in gates:
                gates.remove(QUEUE_GATE)
                t.gated = bool(gates)
                if not t.gated:
                    job._gated = max(0, getattr(job, "_gated", 1) - 1)
                if store is not None:
                    try:
                        store.update("Pod", t.pod)
                    except KeyError:
                        pass
