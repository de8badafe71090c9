"""Backfill action (reference ``actions/backfill/backfill.go:58-120``).

Places BestEffort (zero-request) pending tasks of Inqueue/Running jobs on
any feasible node after allocate — fills fragments.  Reuses the same
score/select kernels: a zero request makes every ready node capacity-
unbounded, so one select pass spreads the class by score.
"""

from __future__ import annotations

from typing import List

import numpy as np

from ...api.types import PodGroupPhase
from ..plan import ClassPlan, CyclePlan, run_plan_hip, run_plan_torch


class BackfillAction:
    name = "backfill"

    def execute(self, ssn) -> None:
        nt = ssn.node_tensors
        if nt is None or nt.n == 0:
            return
        plan = CyclePlan(nt, ssn.queue_limit, ssn.queue_alloc)
        plan.dim_w = ssn.dim_weight_vector()
        plan.bias = getattr(ssn, "score_bias", None)
        predicates = getattr(ssn, "predicates", None)

        from ...api.types import TaskStatus
        # job-order tiers decide who backfills first when pod slots are
        # scarce (reference backfill.go: ssn.JobOrderFn over pending
        # BestEffort owners; tasks within a job by TaskOrder/priority)
        eligible = [j for j in ssn.jobs.values()
                    if j.task_status_index.get(TaskStatus.PENDING)
                    and j.phase in (PodGroupPhase.INQUEUE.value,
                                    PodGroupPhase.RUNNING.value)
                    and ssn.queue_index.get(j.queue) is not None]
        for job in ssn.sorted_jobs(eligible):
            classes: List[ClassPlan] = []
            qi = ssn.queue_index.get(job.queue)
            for tc in sorted(job.pending_classes(),
                             key=lambda c: -c.priority):
                if not tc.tasks[0].best_effort:
                    continue
                # a BestEffort pod still occupies its one-pod slot — use
                # the real (pods-only) request so node pod capacity and
                # queue pod quotas clamp the fill
                req = nt.req_vector(tc.tasks[0])
                if req is None:
                    req = np.zeros(nt.r, dtype=np.float32)
                if predicates is not None:
                    tol, require, forbid = predicates.class_constraints(tc, job)
                else:
                    tol, require, forbid = -1, \
                        np.zeros(max(nt.labels.words, 1), dtype=np.int64), \
                        np.zeros(max(nt.labels.words, 1), dtype=np.int64)
                classes.append(ClassPlan(
                    tclass=tc, job_key=job.key, queue_idx=qi, req=req,
                    tolerated=tol, require=require, forbid=forbid,
                    min_needed=0, w_least=1.0, w_most=0.0, w_bal=0.0))
            if classes:
                # backfill is not gang-gated: min 0 via a shim job desc
                from ..plan import JobPlan
                begin = len(plan.classes)
                plan.classes.extend(classes)
                plan.jobs.append(JobPlan(job_key=job.key, class_begin=begin,
                                         class_end=len(plan.classes),
                                         occupied=0, min_available=0))

        if plan.n_classes == 0:
            return
        plan.finalize()
        use_hip = getattr(ssn.config, "use_hip", False)
        result = (run_plan_hip if use_hip else run_plan_torch)(plan)

        from .allocate import AllocateAction
        AllocateAction._apply(AllocateAction(), ssn, plan, result)
