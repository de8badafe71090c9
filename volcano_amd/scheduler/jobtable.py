"""Columnar per-cycle job worksheet.

Round-2 host profiling: after the node ledger landed, the remaining step
cost was ~10k-job Python property/gate chains repeated by every consumer
(enqueue selection, allocate worksheet build, job ordering, plugin
session opens each re-reading phase/occupied/pending/valid per job).

The JobTable gathers those per-job scalars ONCE per cycle into numpy
columns, so every downstream gate/order/vote is array math:

* static columns (min-resources vector, total-request vector, minAvailable,
  priority, creation time, queue index, task count) are cached ACROSS
  cycles and refreshed per job only when its ``_tver`` bumps (task
  add/remove, podgroup update);
* dynamic columns (phase, occupied, pending, failed) refresh each cycle
  in one tight loop.

Reference analog: the per-cycle maps rebuilt by ``OpenSession``
(/root/reference/pkg/scheduler/framework/session.go:66-165) that actions
then iterate with per-job callback chains
(actions/allocate/allocate.go:142 buildAllocateContext).
"""

from __future__ import annotations

from typing import Dict, List, Optional

import numpy as np

from ..api.types import PodGroupPhase, TaskStatus

# phase encoding
PH_PENDING, PH_INQUEUE, PH_RUNNING, PH_OTHER = 0, 1, 2, 3
_PHASE_MAP = {
    PodGroupPhase.PENDING.value: PH_PENDING,
    PodGroupPhase.INQUEUE.value: PH_INQUEUE,
    PodGroupPhase.RUNNING.value: PH_RUNNING,
}


class JobTable:
    def __init__(self):
        self.jobs: List = []
        self._epoch = -1
        self._vers: Optional[np.ndarray] = None
        self._r = -1
        self._queue_index: Optional[Dict[str, int]] = None
        self._qnames: List[str] = []
        # static columns
        self.minres: Optional[np.ndarray] = None    # [J, R] f64
        self.totreq: Optional[np.ndarray] = None    # [J, R] f64
        self.minav: Optional[np.ndarray] = None     # [J] i64
        self.ntasks: Optional[np.ndarray] = None    # [J] i64
        self.prio: Optional[np.ndarray] = None      # [J] i64
        self.ctime: Optional[np.ndarray] = None     # [J] f64
        self.qi: Optional[np.ndarray] = None        # [J] i64 (-1 unknown)
        self.keys: Optional[np.ndarray] = None      # [J] object (job keys)
        # interned plan-atom signature id (-1: multi-class job).  Used as
        # a tie-break BELOW every plugin order key: jobs the plugins left
        # unordered sort by signature, so identical gangs run back to
        # back and the allocate bundler fuses them (the reference leaves
        # tie order unspecified — sort.Slice is unstable).
        self.sigid: Optional[np.ndarray] = None     # [J] i64
        self._sig_ids: Dict = {}
        # effective gang minimum for atomic jobs: max(minAvailable,
        # minTaskMember[role]) — what the allocate fast path computes
        # per job; here once per static refresh
        self.gangmin: Optional[np.ndarray] = None   # [J] i64
        self.subpol: Optional[np.ndarray] = None    # [J] bool
        # minResources names a resource no dim represents → admission
        # gates must reject (dense projection would silently drop it)
        self.minres_bad: Optional[np.ndarray] = None  # [J] bool
        # dynamic columns (refreshed every cycle)
        self.phase: Optional[np.ndarray] = None     # [J] i8
        self.occ: Optional[np.ndarray] = None       # [J] i64
        self.npend: Optional[np.ndarray] = None     # [J] i64
        self.nfailed: Optional[np.ndarray] = None   # [J] i64

    # -- refresh -------------------------------------------------------------
    def refresh(self, cache, nt, queue_index: Dict[str, int]) -> None:
        jobs_dict = cache.jobs
        epoch = getattr(cache, "jobs_epoch", 0)
        J = len(jobs_dict)
        rebuild = (epoch != self._epoch or J != len(self.jobs)
                   or nt.r != self._r)
        if rebuild:
            self.jobs = list(jobs_dict.values())
            self._epoch = epoch
            self._r = nt.r
            self._build_static(nt)
            self._queue_index = None
        else:
            # per-job static refresh for bumped versions
            vers = self._vers
            dirty = [k for k, j in enumerate(self.jobs)
                     if j._tver != vers[k]]
            if dirty:
                self._refresh_static_rows(dirty, nt)
        if queue_index != self._queue_index:
            self._queue_index = dict(queue_index)
            qidx = self._queue_index
            self.qi = np.fromiter(
                (qidx.get(q, -1) for q in self._qnames),
                dtype=np.int64, count=len(self._qnames))
        self._refresh_dynamic()

    def _build_static(self, nt) -> None:
        jobs = self.jobs
        J, R = len(jobs), nt.r
        self.minres = np.zeros((J, R), dtype=np.float64)
        self.totreq = np.zeros((J, R), dtype=np.float64)
        minav, ntasks, prio, ctime, vers = [], [], [], [], []
        qnames, keys, sigids, gangmin, subpol = [], [], [], [], []
        mrbad = []
        sid = self._sig_ids
        for k, job in enumerate(jobs):
            job._jrow = k
            self.minres[k] = job.minres_vec(nt)
            self.totreq[k] = job.total_req_vec(nt)
            ma = job.min_available
            minav.append(ma)
            ntasks.append(len(job.tasks))
            prio.append(job.priority)
            ctime.append(job.creation_timestamp)
            vers.append(job._tver)
            qnames.append(job.queue)
            keys.append(job.key)
            atom = job.plan_atom()
            if atom:
                sigids.append(sid.setdefault(atom[0], len(sid)))
                mtm = job.min_task_member
                gangmin.append(max(ma, mtm.get(atom[1], 0)) if mtm else ma)
            else:
                sigids.append(-1)
                gangmin.append(ma)
            pg = job.podgroup
            subpol.append(pg is not None and bool(pg.spec.sub_group_policy))
            mrbad.append(job.minres_unrepresented(nt))
        self.minres_bad = np.array(mrbad, dtype=bool)
        self.sigid = np.array(sigids, dtype=np.int64)
        self.gangmin = np.array(gangmin, dtype=np.int64)
        self.subpol = np.array(subpol, dtype=bool)
        self.minav = np.array(minav, dtype=np.int64)
        self.ntasks = np.array(ntasks, dtype=np.int64)
        self.prio = np.array(prio, dtype=np.int64)
        self.ctime = np.array(ctime, dtype=np.float64)
        self._vers = np.array(vers, dtype=np.int64)
        self._qnames = qnames
        self.keys = np.array(keys, dtype=object)

    def _refresh_static_rows(self, rows: List[int], nt) -> None:
        for k in rows:
            job = self.jobs[k]
            self.minres[k] = job.minres_vec(nt)
            self.totreq[k] = job.total_req_vec(nt)
            self.minav[k] = job.min_available
            self.ntasks[k] = len(job.tasks)
            self.prio[k] = job.priority
            self.ctime[k] = job.creation_timestamp
            self._qnames[k] = job.queue
            qidx = self._queue_index
            if qidx is not None:
                self.qi[k] = qidx.get(job.queue, -1)
            atom = job.plan_atom()
            if atom:
                self.sigid[k] = self._sig_ids.setdefault(
                    atom[0], len(self._sig_ids))
                mtm = job.min_task_member
                self.gangmin[k] = max(job.min_available,
                                      mtm.get(atom[1], 0)) if mtm \
                    else job.min_available
            else:
                self.sigid[k] = -1
                self.gangmin[k] = job.min_available
            pg = job.podgroup
            self.subpol[k] = pg is not None and \
                bool(pg.spec.sub_group_policy)
            self.minres_bad[k] = job.minres_unrepresented(nt)
            self._vers[k] = job._tver

    def _refresh_dynamic(self) -> None:
        P, F = TaskStatus.PENDING, TaskStatus.FAILED
        pmap = _PHASE_MAP
        occ, npend, nfailed, phase = [], [], [], []
        for job in self.jobs:
            occ.append(job._occ)
            idx = job.task_status_index
            b = idx.get(P)
            npend.append(len(b) if b else 0)
            b = idx.get(F)
            nfailed.append(len(b) if b else 0)
            pg = job.podgroup
            phase.append(pmap.get(pg.status.phase, PH_OTHER)
                         if pg is not None else PH_PENDING)
        J = len(self.jobs)
        self.occ = np.array(occ, dtype=np.int64) if J else \
            np.zeros(0, dtype=np.int64)
        self.npend = np.array(npend, dtype=np.int64) if J else \
            np.zeros(0, dtype=np.int64)
        self.nfailed = np.array(nfailed, dtype=np.int64) if J else \
            np.zeros(0, dtype=np.int64)
        self.phase = np.array(phase, dtype=np.int8) if J else \
            np.zeros(0, dtype=np.int8)
