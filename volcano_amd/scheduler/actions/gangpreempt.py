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

# ALLOCATED_STATUSES already includes RUNNING — appending it again made
# tasks_with_status yield running victims twice (double eviction)
VICTIM_STATUSES = ALLOCATED_STATUSES
assert TaskStatus.RUNNING in VICTIM_STATUSES


DEFAULT_MAX_DOMAINS = 8     # reference gangpreempt.go:36


class GangPreemptAction:
    name = "gangpreempt"
    same_queue = True

    def __init__(self):
        self.max_domains = DEFAULT_MAX_DOMAINS
        self.allow_whole_bundle = True

    def _parse_arguments(self, ssn) -> None:
        """Action knobs (reference gangpreempt.go:63-77 parseArguments):
        ``maxDomains`` caps the candidate-domain scan (invalid/<=0 falls
        back to the default; domain iteration itself lives in the
        topology preempt path — actions/preempt.py), ``allowWholeBundle``
        toggles whole-job victim bundles."""
        args = getattr(ssn.config, "configurations", {}).get(self.name, {})
        md = args.get("maxDomains")
        try:
            md = int(md) if md is not None else 0
        except (TypeError, ValueError):
            md = 0
        self.max_domains = md if md > 0 else DEFAULT_MAX_DOMAINS
        awb = args.get("allowWholeBundle")
        if isinstance(awb, bool):
            self.allow_whole_bundle = awb
        elif isinstance(awb, str):
            self.allow_whole_bundle = awb.lower() != "false"
        else:
            self.allow_whole_bundle = True

    def execute(self, ssn) -> None:
        nt = ssn.node_tensors
        if nt is None or nt.n == 0 or not ssn.preemptable_fns:
            return
        self._parse_arguments(ssn)
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

    def _victim_bundles(self, ssn, rep, pj: JobInfo, vj: JobInfo):
        """Split a victim job's occupied tasks into a SAFE bundle
        (evictable surplus above the job/role minimums — the victim gang
        survives) and a WHOLE bundle (the remainder; taking it strands
        the gang, so the whole-job semantics of :meth:`_victim_ok`
        apply) — reference ``actions/utils/bundle.go:96-193``
        BuildVictimBundles.  Safe victims pass the FULL plugin
        intersection (including the gang keep-above-min filter: they are
        above-min by construction); whole bundles are gated by
        ``allowWholeBundle`` and the bundle-exempt intersection."""
        victims = vj.tasks_with_status(*VICTIM_STATUSES)
        if not victims or vj.priority >= pj.priority:
            return []
        nt = ssn.node_tensors
        out = []
        surplus = vj.occupied_count - vj.min_available
        mtm = vj.min_task_member or {}
        role_surplus = {r: vj.role_occupied(r) - m for r, m in mtm.items()}
        safe = []
        safe_ids = set()
        for t in sorted(victims, key=lambda t: t.priority):
            rs = role_surplus.get(t.role)
            if surplus > 0 and (rs is None or rs > 0) and t.preemptable:
                safe.append(t)
                safe_ids.add(t.uid)
                surplus -= 1
                if rs is not None:
                    role_surplus[t.role] = rs - 1
        if safe:
            fns = ssn.preemptable_fns if self.same_queue \
                else ssn.reclaimable_fns
            allowed = safe
            for fn in fns:
                keep = {t.uid for t in fn(rep, allowed)}
                allowed = [t for t in allowed if t.uid in keep]
                if not allowed:
                    break
            if allowed:
                gvec = np.zeros(nt.r, dtype=np.float64)
                for t in allowed:
                    v = nt.req_vector(t)
                    if v is not None:
                        gvec += v
                out.append(("safe", vj, allowed, gvec))
                safe_ids = {t.uid for t in allowed}
            else:
                safe_ids = set()
        if self.allow_whole_bundle and self._victim_ok(ssn, rep, pj, vj):
            # disjoint from the safe bundle (evict lists never overlap)
            rest = [t for t in victims if t.uid not in safe_ids]
            if rest:
                gvec = np.zeros(nt.r, dtype=np.float64)
                for t in rest:
                    v = nt.req_vector(t)
                    if v is not None:
                        gvec += v
                out.append(("whole", vj, rest, gvec))
        return out

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

        # candidate victim bundles: SAFE bundles first (no gang cost),
        # then WHOLE, ROI order within kind (lowest priority, fewest
        # evictions) — reference utils.SortBundlesForPreempt
        bundles = []
        for vj in ssn.jobs.values():
            if vj.key == job.key:
                continue
            if self.same_queue and vj.queue != job.queue:
                continue
            if not self.same_queue and vj.queue == job.queue:
                continue
            bundles.extend(self._victim_bundles(ssn, rep, job, vj))
        bundles.sort(key=lambda b: (b[0] != "safe",
                                    b[1].priority, len(b[2])))

        chosen = []
        gain = free.copy()
        for kind, vj, victims, gvec in bundles:
            if (gain + 0.1 >= need_vec).all():
                break
            chosen.append((vj, victims))
            gain += gvec
        if not (gain + 0.1 >= need_vec).all():
            return     # even evicting every bundle would not fit: do nothing

        stmt = Statement(ssn)
        for vj, victims in chosen:
            for v in victims:
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
        self._parse_arguments(ssn)      # gangreclaim shares the knobs
        for q in ssn.sorted_queues():
            if not q.is_open or ssn.queue_overused(q):
                continue
            jobs_in_q = [j for j in ssn.jobs.values() if j.queue == q.name
                         and j.phase in (PodGroupPhase.INQUEUE.value,
                                         PodGroupPhase.RUNNING.value)]
            starving = [j for j in jobs_in_q
                        if ssn.job_starving(j) and j.pending_tasks
                        and ssn.job_valid(j)]
            for job in ssn.sorted_jobs(starving):
                self._gang_preempt(ssn, job)
