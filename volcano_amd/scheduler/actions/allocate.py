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

import time
from typing import Dict, List

import numpy as np

from ...api.info import TaskClass
from ...api.types import PodGroupPhase, TaskStatus

PENDING_S = TaskStatus.PENDING
BOUND_S = TaskStatus.BOUND
from ...utils.metrics import METRICS
from ..plan import (BundleEntry, ClassPlan, CyclePlan, JobPlan, run_plan_hip,
                    run_plan_torch)


class AllocateAction:
    name = "allocate"

    def __init__(self, use_hip: bool = None):
        self.use_hip = use_hip
        # soft-sharding coordinator (parallel/softshard.py) — set by
        # DistributedScheduler when nodes are replicated across ranks
        self.coordinator = None

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
        t0 = time.perf_counter()
        # a plugin may have grown the dim registry (synthetic dims, e.g.
        # interpodaffinity) after the queue tensors were built — pad the
        # queue rows: new dims are queue-unlimited
        if ssn.queue_limit is not None and ssn.queue_limit.shape[1] < nt.r:
            import torch
            Q, old_r = ssn.queue_limit.shape
            pad = nt.r - old_r
            ssn.queue_limit = torch.cat(
                [ssn.queue_limit,
                 torch.full((Q, pad), 1.0e18, dtype=torch.float32)], dim=1)
            ssn.queue_alloc = torch.cat(
                [ssn.queue_alloc, torch.zeros((Q, pad), dtype=torch.float32)],
                dim=1)
        plan = CyclePlan(nt, ssn.queue_limit, ssn.queue_alloc)
        plan.dim_w = ssn.dim_weight_vector()
        plan.bias = getattr(ssn, "score_bias", None)
        w = ssn.score_weights
        predicates = getattr(ssn, "predicates", None)

        # -- build the worksheet (buildAllocateContext analog) --------------
        # Columnar path: selection (phase/pending/valid), queue grouping
        # and job ordering all run as array ops over the JobTable; the
        # legacy per-job path remains for unaligned tables (mid-cycle
        # direct mutations in tests).
        jt = getattr(ssn, "job_table", None)
        if jt is not None and len(jt.jobs) != len(ssn.jobs):
            jt = None
        # parallel worksheets: jobs_l[i] is the i-th job in plan order,
        # qi_l[i] its queue index (no per-job tuple boxing)
        jobs_l: List = []
        qi_l: List[int] = []
        # run metadata (parallel to jobs_l, columnar path only): maximal
        # runs of consecutive atomic same-signature jobs emit as ONE
        # fused bundle without per-job plan_atom/signature work
        elig_l = sig_l = gm_l = None
        if jt is not None:
            from ..jobtable import PH_INQUEUE, PH_RUNNING
            mask = ((jt.phase == PH_INQUEUE) | (jt.phase == PH_RUNNING)) \
                & (jt.npend > 0)
            vcols = ssn.job_valid_cols
            if len(vcols) == len(ssn.job_valid_fns):
                for fn in vcols:
                    mask &= fn(jt)
                sel = np.nonzero(mask)[0]
            else:
                sel = np.nonzero(mask)[0]
                if len(sel):
                    keep = np.fromiter(
                        (ssn.job_valid(jt.jobs[int(k)]) for k in sel),
                        dtype=bool, count=len(sel))
                    sel = sel[keep]
            if len(sel):
                qis = jt.qi[sel]
                qname_by_qi = {i: n for n, i in ssn.queue_index.items()}
                present = np.unique(qis[qis >= 0])
                queues = [ssn.queues[qname_by_qi[int(i)]] for i in present
                          if qname_by_qi.get(int(i)) in ssn.queues]
                jjobs = jt.jobs
                track_runs = not ssn.class_bias_fns
                if track_runs:
                    elig_l, sig_l, gm_l = [], [], []
                for q in ssn.sorted_queues(queues):
                    if not q.is_open or ssn.queue_overused(q):
                        continue
                    # Allocatable is queue-scoped (quota gates; the exact
                    # per-task bound is enforced in-kernel).  CONTRACT:
                    # allocatable_fns must be job-independent (job=None).
                    if not ssn.allocatable(q, None):
                        continue
                    qi = ssn.queue_index[q.name]
                    rows = ssn.ordered_job_rows(jt, sel[qis == qi])
                    # tolist(): native ints, no per-row numpy scalar boxing
                    rl = rows.tolist()
                    jobs_l.extend(jjobs[k] for k in rl)
                    qi_l.extend([qi] * len(rl))
                    if track_runs:
                        elig_l.extend(((jt.sigid[rows] >= 0)
                                       & (jt.npend[rows] == jt.ntasks[rows])
                                       & ~jt.subpol[rows]).tolist())
                        sig_l.extend(jt.sigid[rows].tolist())
                        gm_l.extend(jt.gangmin[rows].tolist())
        else:
            by_queue = {}
            for job in ssn.jobs.values():
                if job.phase not in (PodGroupPhase.INQUEUE.value,
                                     PodGroupPhase.RUNNING.value):
                    continue
                if not job.task_status_index.get(PENDING_S):
                    continue      # no list build — emptiness check only
                if not ssn.job_valid(job):
                    continue
                by_queue.setdefault(job.queue, []).append(job)

            queues = [ssn.queues[q] for q in by_queue if q in ssn.queues]
            for q in ssn.sorted_queues(queues):
                if not q.is_open or ssn.queue_overused(q):
                    continue
                jobs_q = by_queue[q.name]
                gate = ssn.allocatable(q, None) if jobs_q else True
                if not gate:
                    continue
                qi = ssn.queue_index[q.name]
                sj = ssn.sorted_jobs(jobs_q)
                jobs_l.extend(sj)
                qi_l.extend([qi] * len(sj))

        # Gang bundling: a run of consecutive jobs whose single pending
        # class has identical (queue, signature) is fused into ONE kernel
        # pass — ntasks summed, per-job gang minimums enforced at apply
        # (tail-trim).  This is the dominant shape at scale (many identical
        # gangs) and collapses 2 launches/job into 2 launches/run.
        open_bundle: ClassPlan = None
        open_key = None

        def close_bundle():
            nonlocal open_bundle, open_key
            if open_bundle is not None:
                # in-kernel revert only when even the EASIEST bundled gang
                # cannot place (the apply walk settles per-job minimums);
                # min_needed is min-tracked as entries append
                easiest = open_bundle.min_needed
                plan.jobs.append(JobPlan(
                    job_key=open_bundle.job_key,
                    class_begin=len(plan.classes),
                    class_end=len(plan.classes) + 1,
                    occupied=0,
                    min_available=easiest))
                plan.classes.append(open_bundle)
            open_bundle, open_key = None, None

        def bundle_in(cp: ClassPlan, job, qi: int, gang_min: int,
                      sig: str) -> None:
            nonlocal open_bundle, open_key
            key = (qi, sig)
            tasks = cp.tclass.tasks        # already a fresh per-job list
            if open_bundle is not None and key == open_key:
                b = open_bundle
                b.bundle.append(BundleEntry(job.key, tasks, len(tasks),
                                            gang_min, job))
                b.ntasks_override += len(tasks)
                if gang_min < b.min_needed:
                    b.min_needed = gang_min
                return
            close_bundle()
            # open the bundle on the ORIGINAL ClassPlan: the instance
            # count lives in ntasks_override, per-job boundaries in the
            # bundle list; tclass stays the first job's (its tasks[0] is
            # the representative for handlers/constraint builders)
            cp.job_key = job.key
            cp.min_needed = gang_min
            cp.ntasks_override = len(tasks)
            cp.bundle = [BundleEntry(job.key, tasks, len(tasks), gang_min,
                                     job)]
            open_bundle, open_key = cp, key

        _MISS = object()
        req_memo = {}
        # (sig, qi) → (req, tolerated, require, forbid): one lookup per
        # job covers both the request vector and the constraint planes
        cons_memo = {}
        from ...api.types import TaskStatus as _TS
        w_least = w.get("least", 1.0)
        w_most = w.get("most", 0.0)
        w_bal = w.get("bal", 0.0)
        bias_fns = ssn.class_bias_fns

        def class_bias(tc, job):
            out = None
            for fn in bias_fns:
                b = fn(tc, job)
                if b is not None:
                    out = b if out is None else out + b
            return out

        # nominated gangs take the host fast path — exclude them from runs
        if elig_l is not None and self.coordinator is None and \
                getattr(ssn.cache, "had_nominations", False):
            for ix, jb in enumerate(jobs_l):
                if jb._nom:
                    elig_l[ix] = False

        L = len(jobs_l)
        skip_until = -1
        for i in range(L):
            if i < skip_until:
                continue
            job = jobs_l[i]
            qi = qi_l[i]

            # -- fused run: consecutive atomic jobs with one interned
            # signature in one queue emit as a single bundled ClassPlan —
            # no per-job signature/constraint work (the dominant shape
            # after signature-grouped ordering)
            if elig_l is not None and elig_l[i]:
                s = sig_l[i]
                j = i + 1
                while j < L and elig_l[j] and sig_l[j] == s \
                        and qi_l[j] == qi:
                    j += 1
                if j > i + 1:
                    skip_until = j
                    atom = job.plan_atom()
                    sig, role, request, priority = atom
                    rep0 = next(iter(job.task_status_index.get(
                        _TS.PENDING).values()))
                    if rep0.best_effort:
                        continue  # BestEffort is backfill's (allocate.go:265)
                    ckey = (sig, qi)
                    got = cons_memo.get(ckey, _MISS)
                    if got is _MISS:
                        req = nt.req_vector(
                            next(iter(job.task_status_index.get(
                                _TS.PENDING).values())))
                        if req is None:
                            got = None
                        else:
                            tc0 = TaskClass(
                                signature=sig, role=role, request=request,
                                tasks=[next(iter(job.task_status_index.get(
                                    _TS.PENDING).values()))],
                                priority=priority)
                            if predicates is not None:
                                tol, require, forbid = \
                                    predicates.class_constraints(tc0, job)
                            else:
                                tol = -1
                                W = max(nt.labels.words, 1)
                                require = np.zeros(W, dtype=np.int64)
                                forbid = np.zeros(W, dtype=np.int64)
                            got = (req, tol, require, forbid)
                        cons_memo[ckey] = got
                    if got is None:
                        continue  # asks for a resource no node offers
                    req, tol, require, forbid = got
                    entries: List[BundleEntry] = []
                    total = 0
                    mn = 1 << 60
                    for k in range(i, j):
                        jb = jobs_l[k]
                        pend = jb.task_status_index.get(_TS.PENDING)
                        if not pend:
                            continue
                        if next(iter(pend.values())).gated:
                            continue
                        tasks = list(pend.values())
                        gm = gm_l[k]
                        entries.append(BundleEntry(jb.key, tasks,
                                                   len(tasks), gm, jb))
                        total += len(tasks)
                        if gm < mn:
                            mn = gm
                    if not entries:
                        continue
                    close_bundle()
                    e0 = entries[0]
                    tc = TaskClass(signature=sig, role=role,
                                   request=request, tasks=e0.tasks,
                                   priority=priority)
                    cp = ClassPlan(tclass=tc, job_key=e0.job_key,
                                   queue_idx=qi, req=req, tolerated=tol,
                                   require=require, forbid=forbid,
                                   min_needed=mn, w_least=w_least,
                                   w_most=w_most, w_bal=w_bal)
                    cp.ntasks_override = total
                    cp.bundle = entries
                    plan.jobs.append(JobPlan(
                        job_key=e0.job_key,
                        class_begin=len(plan.classes),
                        class_end=len(plan.classes) + 1,
                        occupied=0, min_available=mn))
                    plan.classes.append(cp)
                    continue
                # singleton run: fall through to the per-job path (it
                # may open/continue a bundle with neighbours)

            # -- SubGroupPolicy (reference types.go:218 SubGroupPolicySpec
            # + allocate.go allocateForSubJob): matching pods partition
            # into gang-atomic subgroups of subGroupSize, each planned as
            # its own single-class gang (in-kernel revert); the job-level
            # minSubGroups gate is applied after the run (_apply gating).
            if job.podgroup is not None and \
                    job.podgroup.spec.sub_group_policy:
                close_bundle()
                self._plan_subgroups(ssn, plan, job, qi, predicates,
                                     w_least, w_most, w_bal)
                continue

            # -- nomination fast path (reference allocate.go:797
            # NominatedNodeName): a gang pipelined by last cycle's preempt
            # re-checks fit on its nominated nodes and commits host-side,
            # skipping scoring entirely.  Disabled under soft-sharding
            # (commits must flow through the conflict reconcile).
            if self.coordinator is None and job._nom and \
                    self._try_nominated(ssn, job, qi, predicates):
                continue

            # -- steady-state fast path: the whole job is one pending class
            atom = job.plan_atom()
            pend = job.task_status_index.get(_TS.PENDING, {})
            if atom and pend and len(pend) == len(job.tasks):
                sig, role, request, priority = atom
                first = next(iter(pend.values()))
                if first.gated or first.best_effort:
                    continue    # BestEffort is backfill's (allocate.go:265)
                # bundle continuation: identical (queue, signature) run —
                # append the gang entry directly, skipping the
                # TaskClass/ClassPlan/constraint work bundle_in would
                # discard anyway (dominant shape: thousands of identical
                # gangs back to back)
                if not bias_fns and open_bundle is not None \
                        and open_key == (qi, sig):
                    tasks = list(pend.values())
                    mtm = job._mtm
                    if mtm is None:
                        mtm = job.min_task_member
                    gang_min = job.min_available if not mtm else \
                        max(job.min_available, mtm.get(role, 0))
                    b = open_bundle
                    b.bundle.append(
                        BundleEntry(job.key, tasks, len(tasks), gang_min,
                                    job))
                    b.ntasks_override += len(tasks)
                    if gang_min < b.min_needed:
                        b.min_needed = gang_min
                    continue
                ckey = (sig, qi)
                got = cons_memo.get(ckey, _MISS)
                if got is _MISS:
                    req = nt.req_vector(first)
                    if req is None:
                        got = None
                    else:
                        tc0 = TaskClass(signature=sig, role=role,
                                        request=request, tasks=[first],
                                        priority=priority)
                        if predicates is not None:
                            tol, require, forbid = \
                                predicates.class_constraints(tc0, job)
                        else:
                            tol = -1
                            W = max(nt.labels.words, 1)
                            require = np.zeros(W, dtype=np.int64)
                            forbid = np.zeros(W, dtype=np.int64)
                        got = (req, tol, require, forbid)
                    cons_memo[ckey] = got
                if got is None:
                    continue      # asks for a resource no node offers
                req, tol, require, forbid = got
                tc = TaskClass(signature=sig, role=role, request=request,
                               tasks=list(pend.values()), priority=priority)
                mtm = job._mtm
                if mtm is None:
                    mtm = job.min_task_member
                gang_min = job.min_available if not mtm else \
                    max(job.min_available, mtm.get(role, 0))
                cp = ClassPlan(tclass=tc, job_key=job.key, queue_idx=qi,
                               req=req, tolerated=tol, require=require,
                               forbid=forbid, min_needed=gang_min,
                               w_least=w_least, w_most=w_most, w_bal=w_bal)
                if bias_fns:
                    cp.bias = class_bias(tc, job)
                if cp.bias is not None:     # biased classes never fuse
                    close_bundle()
                    plan.add_job(job, [cp])
                else:
                    bundle_in(cp, job, qi, gang_min, sig)
                continue

            classes: List[ClassPlan] = []
            skipped = False
            for tc in job.pending_classes():
                if tc.tasks[0].best_effort:
                    continue    # BestEffort is backfill's (allocate.go:265)
                req = nt.req_vector(tc.tasks[0])
                if req is None:
                    skipped = True    # asks for a resource no node offers
                    continue
                if predicates is not None:
                    tol, require, forbid = predicates.class_constraints(tc, job)
                else:
                    tol = -1          # tolerate everything
                    W = max(nt.labels.words, 1)
                    require = np.zeros(W, dtype=np.int64)
                    forbid = np.zeros(W, dtype=np.int64)
                need = job.min_task_member.get(tc.role, 0)
                min_needed = max(0, need - job.role_occupied(tc.role))
                cp = ClassPlan(
                    tclass=tc, job_key=job.key, queue_idx=qi, req=req,
                    tolerated=tol, require=require, forbid=forbid,
                    min_needed=min_needed, w_least=w.get("least", 1.0),
                    w_most=w.get("most", 0.0), w_bal=w.get("bal", 0.0))
                if bias_fns:
                    cp.bias = class_bias(tc, job)
                classes.append(cp)
            if skipped and classes and job.min_available > sum(
                    c.tclass.count for c in classes) + job.occupied_count:
                # gang can never be satisfied this cycle
                continue
            if not classes:
                continue

            bundleable = (len(classes) == 1 and job.occupied_count == 0
                          and classes[0].tclass.count == len(job.tasks)
                          and classes[0].bias is None)
            if bundleable:
                cp = classes[0]
                gang_min = max(job.min_available, cp.min_needed)
                bundle_in(cp, job, qi, gang_min, cp.tclass.signature)
            else:
                close_bundle()
                plan.add_job(job, classes)
        close_bundle()

        if plan.n_classes == 0 and self.coordinator is None:
            return
        coord = self.coordinator
        used_before = nt.used_t.clone() if coord is not None else None
        if coord is not None:
            stagger = coord.stagger_bias(nt)
            plan.bias = stagger if plan.bias is None else plan.bias + stagger
        plan.finalize()
        t1 = time.perf_counter()
        result = self._runner(ssn)(plan) if plan.n_classes else None
        t2 = time.perf_counter()
        bad = frozenset()
        if coord is not None:
            bad = coord.find_conflicts(nt, used_before)
        if result is not None:
            self._gate_subgroup_families(plan, result)
            self._apply(ssn, plan, result, bad_nodes=bad)
        if coord is not None:
            coord.finalize(ssn, nt, used_before)
        t3 = time.perf_counter()
        METRICS.observe("allocate:plan_build", t1 - t0)
        METRICS.observe("allocate:plan_run", t2 - t1)
        METRICS.observe("allocate:apply", t3 - t2)

    def _plan_subgroups(self, ssn, plan, job, qi: int, predicates,
                        w_least, w_most, w_bal) -> None:
        """Partition the job's pending pods per SubGroupPolicy entry and
        emit one single-class gang JobPlan per complete subgroup.  Pods
        left unmatched (or in an incomplete tail chunk) stay pending
        this cycle.  Synthetic job keys alias back to the real job for
        the apply walk; the minSubGroups family gate lives in plan
        metadata and is enforced post-run.

        Semantics notes (reference types.go:224: SubGroupPolicy is the
        evolution of minTaskMember): when present it supersedes the
        job-level minMember for admission-to-placement — minSubGroups is
        the job gate; corrective actions (preempt/reclaim) treat the job
        conventionally through its pending tasks."""
        from ...api.types import TaskStatus as _TS
        pend = [t for t in job.task_status_index.get(
            _TS.PENDING, {}).values() if not t.gated]
        claimed = set()
        if not hasattr(plan, "job_alias"):
            plan.job_alias = {}
            plan.subgroup_families = []
        for pi, pol in enumerate(job.podgroup.spec.sub_group_policy):
            size = max(1, int(pol.get("subGroupSize", 1)))
            min_subs = int(pol.get("minSubGroups", 0))
            sel = pol.get("labelSelector") or {}
            keys = pol.get("matchLabelKeys") or []
            topo = pol.get("networkTopology")
            matched = []
            for t in pend:
                if id(t) in claimed or t.pod is None:
                    continue
                lbl = t.pod.meta.labels
                if all(lbl.get(k) == v for k, v in sel.items()):
                    matched.append(t)
            groups: Dict[tuple, List] = {}
            for t in matched:
                gk = tuple(t.pod.meta.labels.get(k) for k in keys)
                groups.setdefault(gk, []).append(t)
            members = []
            sg = 0
            for gk in sorted(groups, key=lambda g: tuple(map(str, g))):
                ts = groups[gk]
                for off in range(0, len(ts) - size + 1, size):
                    chunk = ts[off:off + size]
                    for t in chunk:
                        claimed.add(id(t))
                    first = chunk[0]
                    req = ssn.node_tensors.req_vector(first)
                    if req is None:
                        continue
                    tc = TaskClass(signature=first.class_signature(),
                                   role=first.role, request=first.request,
                                   tasks=chunk, priority=first.priority,
                                   topology=topo)
                    if predicates is not None:
                        tol, require, forbid = \
                            predicates.class_constraints(tc, job)
                    else:
                        W = max(ssn.node_tensors.labels.words, 1)
                        tol = -1
                        require = np.zeros(W, dtype=np.int64)
                        forbid = np.zeros(W, dtype=np.int64)
                    skey = f"{job.key}#sg{pi}:{sg}"
                    sg += 1
                    cp = ClassPlan(tclass=tc, job_key=skey, queue_idx=qi,
                                   req=req, tolerated=tol, require=require,
                                   forbid=forbid, min_needed=size,
                                   w_least=w_least, w_most=w_most,
                                   w_bal=w_bal)
                    plan.jobs.append(JobPlan(
                        job_key=skey, class_begin=len(plan.classes),
                        class_end=len(plan.classes) + 1, occupied=0,
                        min_available=size))
                    plan.classes.append(cp)
                    plan.job_alias[skey] = job.key
                    members.append(skey)
            if members:
                plan.subgroup_families.append((job.key, min_subs, members))

    def _gate_subgroup_families(self, plan, result) -> None:
        """minSubGroups: if fewer than the required number of a policy's
        subgroups fully placed, revert the ones that did (whole-family
        atomicity above the per-subgroup gangs).  Mutates the ARRAY
        result (job_flag / log_counts) — the apply path reads those."""
        families = getattr(plan, "subgroup_families", [])
        if not families:
            return
        class_of = {cp.job_key: c for c, cp in enumerate(plan.classes)}
        jf = result.job_flag
        cls_job = plan.class_job
        for real_key, min_subs, members in families:
            placed = [k for k in members
                      if k in class_of
                      and jf[cls_job[class_of[k]]]
                      and result.class_placed[class_of[k]] > 0]
            if len(placed) >= min_subs:
                continue
            for k in placed:
                c = class_of[k]
                cp = plan.classes[c]
                ln, lc = result.class_entries(c)
                self._revert_pieces(
                    plan, cp,
                    [(int(n), int(cnt)) for n, cnt in zip(ln, lc)
                     if cnt > 0])
                lc[:] = 0
                result.class_placed[c] = 0
                jf[cls_job[c]] = 0

    def _try_nominated(self, ssn, job, qi: int, predicates) -> bool:
        """Commit a fully-nominated gang onto its nominated nodes if they
        still fit (capacity + ready + taints + label planes re-checked on
        the packed mirrors).  Any miss clears the nominations and returns
        False — the job takes the normal scored path."""
        from ...api.types import TaskStatus as _TS
        job._nom = False      # hint is single-shot per cycle
        pend_idx = job.task_status_index.get(_TS.PENDING)
        if not pend_idx:
            return False
        if not next(iter(pend_idx.values())).nominated_node:
            return False    # common case: no nomination — zero-alloc exit
        import torch
        from ...api.resource import Resource
        pend = list(pend_idx.values())
        nt = ssn.node_tensors
        by_node: Dict[str, List] = {}
        for t in pend:
            ni = ssn.nodes.get(t.nominated_node)
            if t.gated or ni is None or ni.node_id < 0:
                self._clear_nominations(pend)
                return False
            by_node.setdefault(t.nominated_node, []).append(t)

        # constraint planes per class signature (memoized by predicates)
        sig_cons = {}
        total_vec = np.zeros(nt.r, dtype=np.float64)
        for nn, ts in by_node.items():
            ni = ssn.nodes[nn]
            nid = ni.node_id
            if not nt.ready_np[nid]:
                self._clear_nominations(pend)
                return False
            need = Resource()
            for t in ts:
                need.add(t.request)
                if predicates is not None:
                    sig = t.class_signature()
                    cons = sig_cons.get(sig)
                    if cons is None:
                        tc = TaskClass(signature=sig, role=t.role,
                                       request=t.request, tasks=[t],
                                       priority=t.priority)
                        cons = sig_cons[sig] = \
                            predicates.class_constraints(tc, job)
                    tol, require, forbid = cons
                    if (nt.taints_np[nid] & ~np.int64(tol)) != 0:
                        self._clear_nominations(pend)
                        return False
                    col = nt.planes_np[:, nid]
                    W = min(len(require), len(col))
                    if (require[:W] & ~col[:W]).any() or require[W:].any() \
                            or (forbid[:W] & col[:W]).any():
                        self._clear_nominations(pend)
                        return False
            if not need.less_equal(ni.idle):
                self._clear_nominations(pend)
                return False
            total_vec += nt.resource_vector(need)

        # queue bound (the in-kernel clamp's host equivalent)
        if ssn.queue_limit is not None:
            head = ssn.queue_alloc[qi].numpy().astype(np.float64) + total_vec
            lim = ssn.queue_limit[qi].numpy()
            if not bool((head <= lim + 1e-3).all()):
                self._clear_nominations(pend)
                return False

        # -- commit (host infos + device planes + queue rows) ---------------
        dev = nt.used_t.device
        ids, vecs = [], []
        to_bind = []
        for nn, ts in by_node.items():
            ni = ssn.nodes[nn]
            by_sig: Dict[object, List] = {}
            for t in ts:
                t.node_name = nn
                by_sig.setdefault(t.class_signature(), []).append(t)
            node_need = Resource()
            for group in by_sig.values():
                ni.add_allocated_bulk(group, group[0].request, len(group))
                g0 = group[0]
                tc = TaskClass(signature=g0.class_signature(), role=g0.role,
                               request=g0.request, tasks=group,
                               priority=g0.priority)
                ssn.fire_allocate(tc, [ni.node_id], [len(group)], group)
                for t in group:
                    node_need.add(t.request)
            ids.append(ni.node_id)
            vecs.append(nt.resource_vector(node_need))
            to_bind.extend(ts)
        delta = torch.from_numpy(
            np.stack(vecs).astype(np.float32).T).to(dev)      # [R, k]
        idx = torch.tensor(ids, dtype=torch.long, device=dev)
        nt.used_t.index_add_(1, idx, delta)
        if ssn.queue_alloc is not None:
            ssn.queue_alloc[qi] += torch.from_numpy(
                total_vec.astype(np.float32))
        self._clear_nominations(pend)
        ssn.cache.bind_tasks(to_bind, by_job={job.key: to_bind})
        if ssn.job_ready(job) and job.podgroup is not None and \
                job.phase != PodGroupPhase.RUNNING.value:
            job.podgroup.status.phase = PodGroupPhase.RUNNING.value
            ssn.cache.update_podgroup(job)
        METRICS.inc("allocate:nominated_fastpath")
        return True

    @staticmethod
    def _clear_nominations(tasks) -> None:
        for t in tasks:
            t.nominated_node = ""

    # -- statement commit (host mirror of the device-side state) ------------
    def _apply(self, ssn, plan: CyclePlan, result,
               bad_nodes: frozenset = frozenset()) -> None:
        """Apply the cycle result.  ``bad_nodes`` (soft-sharding conflict
        resolution) lists node ids this rank LOST in the cross-rank
        reconcile: any job with a placement there is reverted wholesale
        (gang atomicity) and its device usage unwound."""
        nt = ssn.node_tensors
        nodes_sorted = getattr(ssn.cache, "nodes_sorted", None)
        if nodes_sorted is None or len(nodes_sorted) != len(ssn.nodes):
            nodes_sorted = sorted(ssn.nodes.values(), key=lambda n: n.name)
        ledger = getattr(ssn.cache, "ledger", None)

        bind_by_job: Dict[str, List] = {}
        committed_jobs: Dict[str, object] = {}     # key -> JobInfo
        full_keys = set()       # jobs whose ENTIRE task set placed —
        # gang-ready by construction, no is_ready() walk needed
        # fire event handlers only when someone registered one — building
        # the per-piece argument lists for nobody was measurable at 10k jobs
        fire = ssn.fire_allocate if ssn.event_handlers else None
        jf = result.job_flag
        cls_job = plan.class_job
        ln_all = result.log_nodes
        lc_all = result.log_counts
        ll = result.log_len
        # per-piece ledger accumulation across ALL classes — one
        # np.add.at after the walk (was one bulk op per class)
        acc_rows: List[int] = []
        acc_cnts: List[int] = []
        acc_cls: List[int] = []

        def commit_pieces(job, cp, c, pieces):
            """Assign (node_id, count) pieces to the next tasks of `job`.
            Node accounting accumulates per piece for the single end-of-
            apply np.add.at; node→task membership defers into
            NodeInfo._batches (folded lazily by cold readers).  Status
            moves once, PENDING→BOUND, inside bind_tasks."""
            jl = bind_by_job.setdefault(job.key, [])
            for nid, count, tasks in pieces:
                ni = nodes_sorted[nid]
                name = ni.name
                for t in tasks:
                    t.node_name = name
                    t.status = BOUND_S    # index repaired by finish_bind
                ni._batches.append(tasks)
                acc_rows.append(nid)
                acc_cnts.append(count)
                acc_cls.append(c)
                jl.extend(tasks)
            committed_jobs[job.key] = job

        for c, cp in enumerate(plan.classes):
            if not jf[cls_job[c]]:
                continue
            n = int(ll[c])
            if n == 0:
                continue
            off = cp.log_off
            if cp.bundle is None:
                job = ssn.jobs.get(cp.job_key) or \
                    ssn.jobs[getattr(plan, "job_alias", {})[cp.job_key]]
                tasks = iter(cp.tclass.tasks)
                pieces = []
                hit_bad = False
                for e in range(off, off + n):
                    cnt = int(lc_all[e])
                    if cnt <= 0:
                        continue
                    nid = int(ln_all[e])
                    if bad_nodes and nid in bad_nodes:
                        hit_bad = True
                    pieces.append((nid, cnt,
                                   [next(tasks) for _ in range(cnt)]))
                if not pieces:
                    continue
                if hit_bad:
                    self._revert_pieces(plan, cp,
                                        [(nid, cnt) for nid, cnt, _ in pieces])
                    continue
                commit_pieces(job, cp, c, pieces)
                if fire is not None:
                    fire(cp.tclass,
                         [nid for nid, _, _ in pieces],
                         [cnt for _, cnt, _ in pieces],
                         [t for _, _, ts in pieces for t in ts])
            else:
                # -- bundle: walk jobs over the placement stream.  An entry
                # that misses its gang minimum releases its slots back to
                # the pool (cursor reset) so smaller jobs behind it still
                # place; an entry touching a lost node (soft-shard
                # conflict) reverts its slots outright.  Unclaimed slots
                # revert at the end.
                ei, eoff = 0, 0
                reverted: List[tuple] = []
                for be in cp.bundle:
                    ei0, eoff0 = ei, eoff
                    pieces = []
                    need = be.ntasks
                    toff = 0
                    while need > 0 and ei < n:
                        nid = int(ln_all[off + ei])
                        cnt = int(lc_all[off + ei])
                        avail = cnt - eoff
                        take = avail if avail < need else need
                        pieces.append((nid, take, be.tasks[toff:toff + take]))
                        toff += take
                        need -= take
                        eoff += take
                        if eoff == cnt:
                            ei += 1
                            eoff = 0
                    if be.ntasks - need < be.min_needed:
                        ei, eoff = ei0, eoff0      # recycle the slots
                        continue
                    if bad_nodes and any(nid in bad_nodes
                                         for nid, _, _ in pieces):
                        reverted.extend((nid, cnt) for nid, cnt, _ in pieces)
                        continue
                    job = be.job if be.job is not None \
                        else ssn.jobs[be.job_key]
                    commit_pieces(job, cp, c, pieces)
                    if need == 0:
                        # bundle entries are whole-job (all tasks were
                        # pending): zero remainder ⇒ fully placed
                        full_keys.add(be.job_key)
                    if fire is not None:
                        fire(cp.tclass, [p[0] for p in pieces],
                             [p[1] for p in pieces],
                             [t for _, _, ts in pieces for t in ts])
                if ei < n:
                    # slots the walk never claimed
                    if eoff:
                        reverted.append((int(ln_all[off + ei]),
                                         int(lc_all[off + ei]) - eoff))
                        ei += 1
                    reverted.extend((int(ln_all[off + e]),
                                     int(lc_all[off + e]))
                                    for e in range(ei, n))
                if reverted:
                    self._revert_pieces(plan, cp, reverted)

        if acc_rows:
            rows = np.asarray(acc_rows, dtype=np.int64)
            cnts = np.asarray(acc_cnts, dtype=np.float64)
            vals = cnts[:, None] * plan.req_np[
                np.asarray(acc_cls, dtype=np.int64)].astype(np.float64)
            if ledger is not None:
                ledger.add_used_rows(rows, vals)
            else:   # bare-cache tests: per-piece Resource math
                for k in range(len(acc_rows)):
                    cp = plan.classes[acc_cls[k]]
                    nodes_sorted[acc_rows[k]]._acct(
                        cp.tclass.request.clone().multi(float(acc_cnts[k])),
                        1, 0, 0)

        if bind_by_job:
            ssn.cache.bind_tasks(None, by_job=bind_by_job, preset=True)

        # flip gang-ready podgroups to Running (job_updater analog) —
        # walk the committed set directly (was a re-walk of every plan
        # class/bundle entry)
        alias = getattr(plan, "job_alias", {})
        jr = ssn.job_ready_fns
        fast_ready = len(jr) == 1 and getattr(jr[0], "is_gang", False)
        running = PodGroupPhase.RUNNING.value
        store = getattr(ssn.cache, "store", None)
        seen = set()
        for key, job in committed_jobs.items():
            real = alias.get(key)
            if real is not None:
                if real in seen:
                    continue
                seen.add(real)
                job = ssn.jobs[real]
            if fast_ready and key in full_keys:
                ready = True        # whole gang placed this cycle
            else:
                ready = (job.is_ready() and job.roles_ready()) \
                    if fast_ready else ssn.job_ready(job)
            pg = job.podgroup
            if ready and pg is not None and pg.status.phase != running:
                pg.status.phase = running
                if store is not None:
                    ssn.cache.update_podgroup(job)

    @staticmethod
    def _revert_pieces(plan: CyclePlan, cp, pieces) -> None:
        """Unwind device-side staging for rejected (node_id, count) pieces
        of one class (gang-unmet tails, soft-shard conflict losers).
        index_add_ accumulates, so duplicate node ids are fine."""
        import torch
        if not pieces:
            return
        nt = plan.nt
        dev = nt.used_t.device
        req = torch.from_numpy(cp.req).to(dev)
        idx = torch.tensor([nid for nid, _ in pieces], dtype=torch.long,
                           device=dev)
        cvec = torch.tensor([float(c) for _, c in pieces],
                            dtype=torch.float32, device=dev)
        nt.used_t.index_add_(1, idx, -req.unsqueeze(1) * cvec.unsqueeze(0))
        total = sum(c for _, c in pieces)
        plan.queue_alloc[cp.queue_idx] -= float(total) * torch.from_numpy(cp.req)
