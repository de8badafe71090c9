"""Allocate action — the core gang allocator.

Reference: ``actions/allocate/allocate.go:122-953`` (buildAllocateContext →
per queue/job: predicate → prioritize → select → Statement.Allocate →
gang commit/Discard).  MI355X redesign: the host builds a *cycle plan* —
queues ordered by the queue-order tier, jobs by the job-order tier, one
ClassPlan per batch of identical pending tasks — and the whole plan runs
as one kernel sequence on the GPU (``plan.run_plan_hip``; CPU oracle
``run_plan_torch``).  Gang commit/rollback happens ON DEVICE (statement
semantics in-kernel); the single readback yields every placement, which
this action applies to the host infos and hands to the bind pipeline.

Known order deviation vs the reference (documented, bounded): the
reference re-sorts queues by share after every job; the plan freezes the
order at cycle start.  Within a cycle this can change which of two
same-share queues goes first; across cycles shares converge identically.
"""

from __future__ import annotations

from typing import List

import numpy as np

from ...api.types import PodGroupPhase, TaskStatus
from ..plan import ClassPlan, CyclePlan, run_plan_hip, run_plan_torch


class AllocateAction:
    name = "allocate"

    def __init__(self, use_hip: bool = None):
        self.use_hip = use_hip

    def _runner(self, ssn):
        if self.use_hip is None:
            use = getattr(ssn.config, "use_hip", False)
        else:
            use = self.use_hip
        return run_plan_hip if use else run_plan_torch

    def execute(self, ssn) -> None:
        nt = ssn.node_tensors
        if nt is None or nt.n == 0:
            return
        plan = CyclePlan(nt, ssn.queue_limit, ssn.queue_alloc)
        plan.dim_w = ssn.dim_weight_vector()
        w = ssn.score_weights
        predicates = getattr(ssn, "predicates", None)

        # -- build the worksheet (buildAllocateContext analog) --------------
        by_queue = {}
        for job in ssn.jobs.values():
            if job.phase not in (PodGroupPhase.INQUEUE.value,
                                 PodGroupPhase.RUNNING.value):
                continue
            if not job.pending_tasks:
                continue
            if not ssn.job_valid(job):
                continue
            by_queue.setdefault(job.queue, []).append(job)

        queues = [ssn.queues[q] for q in by_queue if q in ssn.queues]
        ordered_jobs = []
        for q in ssn.sorted_queues(queues):
            if not q.is_open or ssn.queue_overused(q):
                continue
            for job in ssn.sorted_jobs(by_queue[q.name]):
                if not ssn.allocatable(q, job):
                    continue
                ordered_jobs.append((q, job))

        for q, job in ordered_jobs:
            qi = ssn.queue_index[q.name]
            classes: List[ClassPlan] = []
            skipped = False
            for tc in job.pending_classes():
                req = nt.req_vector(tc.tasks[0])
                if req is None:
                    skipped = True    # asks for a resource no node offers
                    continue
                if predicates is not None:
                    tol, require, forbid = predicates.class_constraints(tc)
                else:
                    tol = -1          # tolerate everything
                    W = max(nt.labels.words, 1)
                    require = np.zeros(W, dtype=np.int64)
                    forbid = np.zeros(W, dtype=np.int64)
                need = job.min_task_member.get(tc.role, 0)
                min_needed = max(0, need - job.role_occupied(tc.role))
                classes.append(ClassPlan(
                    tclass=tc, job_key=job.key, queue_idx=qi, req=req,
                    tolerated=tol, require=require, forbid=forbid,
                    min_needed=min_needed, w_least=w.get("least", 1.0),
                    w_most=w.get("most", 0.0), w_bal=w.get("bal", 0.0)))
            if skipped and classes and job.min_available > sum(
                    c.tclass.count for c in classes) + job.occupied_count:
                # gang can never be satisfied this cycle
                continue
            plan.add_job(job, classes)

        if plan.n_classes == 0:
            return
        plan.finalize()
        result = self._runner(ssn)(plan)
        self._apply(ssn, plan, result)

    # -- statement commit (host mirror of the device-side state) ------------
    def _apply(self, ssn, plan: CyclePlan, result) -> None:
        node_by_id = {}
        nodes_sorted = sorted(ssn.nodes.values(), key=lambda n: n.name)
        for i, ni in enumerate(nodes_sorted):
            node_by_id[i] = ni

        to_bind = []
        for cp, cres in zip(plan.classes, result.class_results):
            if not result.job_committed.get(cp.job_key, False):
                continue
            if not cres.placements:
                continue
            job = ssn.jobs[cp.job_key]
            tasks = iter(cp.tclass.tasks)
            node_ids = [nid for nid, _ in cres.placements]
            counts = [c for _, c in cres.placements]
            for nid, count in cres.placements:
                ni = node_by_id[nid]
                for _ in range(count):
                    task = next(tasks)
                    task.node_name = ni.name
                    job.update_task_status(task, TaskStatus.ALLOCATED)
                    ni.add_task(task)
                    to_bind.append(task)
            ssn.fire_allocate(cp.tclass, node_ids, counts)

        if to_bind:
            ssn.cache.bind_tasks(to_bind)

        # flip gang-ready podgroups to Running (job_updater analog)
        for jp in plan.jobs:
            job = ssn.jobs[jp.job_key]
            if result.job_committed.get(jp.job_key) and ssn.job_ready(job):
                if job.podgroup is not None and \
                        job.phase != PodGroupPhase.RUNNING.value:
                    job.podgroup.status.phase = PodGroupPhase.RUNNING.value
                    ssn.cache.update_podgroup(job)
