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
        by_queue = {}
        for job in ssn.jobs.values():
            if job.phase != PodGroupPhase.PENDING.value:
                continue
            by_queue.setdefault(job.queue, []).append(job)
        queues = [ssn.queues[q] for q in by_queue if q in ssn.queues]
        for q in ssn.sorted_queues(queues):
            if not q.is_open:
                continue
            fifo = q.queue.spec.dequeue_strategy == "fifo"
            jobs_q = by_queue[q.name]
            # bulk fast path: when every enqueueable plugin can admit the
            # whole batch (total demand fits), skip the per-job votes —
            # identical outcome for monotone-sum admission
            bulk = ssn.job_enqueueable_bulk_fns
            if bulk and len(bulk) == len(ssn.job_enqueueable_fns):
                # each fn returns a commit thunk iff the whole batch fits;
                # commit only once EVERY plugin agreed (no partial updates)
                commits = [fn(q.name, jobs_q) for fn in bulk]
                if all(c is not None for c in commits):
                    for c in commits:
                        c()
                    for job in jobs_q:
                        if job.podgroup is not None:
                            job.podgroup.status.phase = \
                                PodGroupPhase.INQUEUE.value
                        ssn.cache.update_podgroup(job)
                        self._lift_queue_gates(ssn, job)
                    continue
            for job in ssn.sorted_jobs(jobs_q):
                if ssn.job_enqueueable(job):
                    if job.podgroup is not None:
                        job.podgroup.status.phase = PodGroupPhase.INQUEUE.value
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
