"""Preempt action (reference ``actions/preempt/preempt.go:101-673``).

Intra-queue priority preemption for starving jobs: per starving gang, per
pending task — find a node where evicting lower-priority victims frees
enough room, evict just-enough victims (victim set = tier-intersection of
the Preemptable callbacks: priority < preemptor, gang keeps victims' jobs
at minAvailable, DRF share, conformance), Pipeline the preemptor onto the
node; commit only if the gang reaches pipelined state, else reverse-order
Discard (Statement).

Preemption is a rare corrective path (runs only for starving jobs after
allocate), so it stays host-side; the per-node feasibility scan uses the
packed dense vectors, not per-pod Resource maps.
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from ...api.info import JobInfo, NodeInfo, TaskInfo
from ...api.types import ALLOCATED_STATUSES, PodGroupPhase, TaskStatus
from ..statement import Statement


def victim_sort_key(ssn, victim: TaskInfo):
    """Lowest-value victims first: priority asc, youngest first
    (reference victim ordering, session_plugins victim order)."""
    job = ssn.jobs.get(victim.job_key)
    created = job.creation_timestamp if job else 0.0
    return (victim.priority, -created)


def _node_fits(nt, ni: NodeInfo, req_vec: np.ndarray, extra: float = 0.1) -> bool:
    fi = ni.future_idle
    for name, i in nt.dims.index.items():
        r = req_vec[i]
        if r > 0.1 and fi.get(name) + extra < r:
            return False
    return True


def _class_feasible_on_node(ssn, cp_constraints, task: TaskInfo,
                            ni: NodeInfo) -> bool:
    nt = ssn.node_tensors
    tol, require, forbid = cp_constraints
    nid = ni.node_id
    if not ni.ready:
        return False
    if int(nt.taints_np[nid]) & ~tol:
        return False
    for w in range(len(require)):
        p = int(nt.planes_np[w, nid])
        if (p & int(require[w])) != int(require[w]) or (p & int(forbid[w])):
            return False
    return True


class PreemptAction:
    name = "preempt"

    #: victims eligible: running/bound tasks that are preemptable targets
    victim_statuses = ALLOCATED_STATUSES + (TaskStatus.RUNNING,)

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
                self._preempt_for_job(ssn, job, same_queue=True)

    # shared with reclaim (cross-queue variant)
    def _preempt_for_job(self, ssn, job: JobInfo, same_queue: bool,
                         victim_filter=None) -> None:
        nt = ssn.node_tensors
        stmt = Statement(ssn)
        predicates = getattr(ssn, "predicates", None)
        still_needed = job.min_available - job.occupied_count - job.waiting_count
        if still_needed <= 0:
            return

        placed = 0
        for tc in job.pending_classes():
            if placed >= still_needed:
                break
            req_vec = nt.req_vector(tc.tasks[0])
            if req_vec is None:
                continue
            constraints = predicates.class_constraints(tc, job) if predicates \
                else (-1, np.zeros(max(nt.labels.words, 1), dtype=np.int64),
                      np.zeros(max(nt.labels.words, 1), dtype=np.int64))
            for task in tc.tasks:
                if placed >= still_needed:
                    break
                node = self._preempt_one(ssn, stmt, job, task, tc, req_vec,
                                         constraints, same_queue,
                                         victim_filter)
                if node is not None:
                    placed += 1

        if ssn.job_pipelined(job) and job.is_pipelined():
            stmt.commit()
        else:
            stmt.discard()

    def _preempt_one(self, ssn, stmt: Statement, job: JobInfo,
                     task: TaskInfo, tc, req_vec, constraints,
                     same_queue: bool, victim_filter) -> Optional[str]:
        nt = ssn.node_tensors
        best: Optional[NodeInfo] = None
        best_victims: List[TaskInfo] = []

        for ni in ssn.nodes.values():
            if not _class_feasible_on_node(ssn, constraints, task, ni):
                continue
            if _node_fits(nt, ni, req_vec):
                # free room already (released by a previous eviction)
                best, best_victims = ni, []
                break
            # candidate victims on this node
            cands = [t for t in ni.tasks.values()
                     if t.status in self.victim_statuses]
            if same_queue:
                cands = [t for t in cands
                         if ssn.jobs.get(t.job_key) is not None
                         and ssn.jobs[t.job_key].queue == job.queue]
            if victim_filter is not None:
                cands = victim_filter(task, cands)
            else:
                cands = ssn.preemptable(task, cands)
            if not cands:
                continue
            cands.sort(key=lambda v: victim_sort_key(ssn, v))
            # evict-just-enough simulation
            chosen = []
            fi = ni.future_idle
            avail = {k: fi.get(k) for k in nt.dims.index}
            def fits():
                for name, i in nt.dims.index.items():
                    if req_vec[i] > 0.1 and avail[name] + 0.1 < req_vec[i]:
                        return False
                return True
            for v in cands:
                if fits():
                    break
                chosen.append(v)
                for name, val in v.request.q.items():
                    if name in avail:
                        avail[name] += val
            if not fits():
                continue
            if best is None or len(chosen) < len(best_victims):
                best, best_victims = ni, chosen
                if not chosen:
                    break

        if best is None:
            return None
        for v in best_victims:
            stmt.evict(v)
        stmt.pipeline(task, best.name)
        return best.name
