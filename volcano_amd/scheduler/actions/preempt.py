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

from typing import Optional

import numpy as np

from ...api.info import JobInfo, NodeInfo, TaskInfo
from ...api.types import ALLOCATED_STATUSES, PodGroupPhase, TaskStatus
from ..statement import Statement


def victim_sort_key(ssn, victim: TaskInfo):
    """Lowest-value victims first: victim queue priority asc (reclaim
    gives back from the least important queue first — reference
    reclaim_test.go "different queue priority"), then task priority asc,
    youngest first."""
    job = ssn.jobs.get(victim.job_key)
    created = job.creation_timestamp if job else 0.0
    q = ssn.queues.get(job.queue) if job else None
    qprio = q.priority if q is not None else 0
    return (qprio, victim.priority, -created)


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
    victim_statuses = ALLOCATED_STATUSES    # includes RUNNING

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
            scan_memo = {}   # per-cycle: identical preemptor shapes
            for job in ssn.sorted_jobs(starving):
                # hard-network-topology gangs preempt INTO a domain: the
                # whole gang must land in one hypernode, so victims are
                # selected per candidate domain via dry-run simulation
                # (reference preempt.go:479 topologyAwarePreempt +
                # actions/utils/simulate.go:40 BuildNominationPlanInDomain)
                if self._job_hard_topology(ssn, job):
                    if self._topology_preempt_for_job(ssn, job,
                                                      same_queue=True):
                        continue
                self._preempt_for_job(ssn, job, same_queue=True,
                                      memo=scan_memo)

    @staticmethod
    def _job_hard_topology(ssn, job: JobInfo) -> bool:
        if getattr(ssn, "hypernode_tree", None) is None:
            return False
        spec = job.podgroup.spec.network_topology if job.podgroup else None
        return bool(spec) and spec.get("mode", "hard") == "hard"

    def _topology_preempt_for_job(self, ssn, job: JobInfo, same_queue: bool,
                                  victim_filter=None) -> bool:
        """Domain-gradient preemption: walk candidate hypernodes in tier
        order (closest domains first — reference HyperNodeGradient), and
        for each run a DRY-RUN trial (Statement applied to session state,
        discarded on failure): evict just-enough in-domain victims, then
        pipeline the gang's pending tasks onto in-domain nodes.  The
        first domain whose trial pipelines the whole gang wins; its
        statement commits and the pipelined tasks carry their node
        nominations into the next cycle (cache._demote_pipelined →
        allocate._try_nominated).  Mirrors preempt.go:479-673 +
        simulate.go:40-300 runSimulateTrialAtHyperNode on the plan
        machinery."""
        tree = ssn.hypernode_tree
        spec = job.podgroup.spec.network_topology
        max_tier = spec.get("highestTierAllowed")
        running_nodes = {t.node_name for t in job.tasks.values()
                         if t.node_name}
        for hname in tree.domains_by_tier(max_tier):
            members = tree.members[hname]
            if running_nodes and not running_nodes <= members:
                continue   # gang fragments must share the domain
            if self._trial_in_domain(ssn, job, members, same_queue,
                                     victim_filter):
                return True
        return False

    def _trial_in_domain(self, ssn, job: JobInfo, members, same_queue: bool,
                         victim_filter) -> bool:
        """One simulate trial: evictions + pipelines applied to session
        state through a Statement; commit iff the gang reaches pipelined
        state inside THIS domain, else reverse-order discard (the
        reference's tmpStmt merge/discard, preempt.go:505-519)."""
        nt = ssn.node_tensors
        predicates = getattr(ssn, "predicates", None)
        stmt = Statement(ssn)
        still_needed = job.min_available - job.occupied_count \
            - job.waiting_count
        if still_needed <= 0:
            return False
        placed = 0
        for tc in job.pending_classes():
            if placed >= still_needed:
                break
            rep0 = tc.tasks[0]
            if rep0.pod is not None and \
                    rep0.pod.preemption_policy == "Never":
                continue    # reference preempt.go:441 (#3642): a Never
                # preemptor may wait but must not displace anything
            req_vec = nt.req_vector(rep0)
            if req_vec is None:
                continue
            constraints = predicates.class_constraints(
                tc, job, skip_topology=True) if predicates \
                else (-1, np.zeros(max(nt.labels.words, 1), dtype=np.int64),
                      np.zeros(max(nt.labels.words, 1), dtype=np.int64))
            scan = self._candidate_scan(ssn, job, tc, req_vec, constraints,
                                        same_queue, victim_filter,
                                        node_filter=members)
            if scan is None:
                continue
            order, victims_by_node = scan
            for task in tc.tasks:
                if placed >= still_needed:
                    break
                node = self._preempt_one(ssn, stmt, job, task, req_vec,
                                         order, victims_by_node)
                if node is not None:
                    placed += 1
        if placed >= still_needed and ssn.job_pipelined(job) \
                and job.is_pipelined():
            stmt.commit()
            return True
        stmt.discard()
        return False

    # shared with reclaim (cross-queue variant)
    def _preempt_for_job(self, ssn, job: JobInfo, same_queue: bool,
                         victim_filter=None, memo=None) -> None:
        nt = ssn.node_tensors
        stmt = Statement(ssn)
        predicates = getattr(ssn, "predicates", None)
        still_needed = job.min_available - job.occupied_count - job.waiting_count
        if still_needed <= 0:
            return

        # queue-quota headroom (reference #3161 cases: a queue AT its
        # capability must evict its own lower-priority work to admit the
        # preemptor — node-level fit alone is not admission)
        headroom = None
        qi = ssn.queue_index.get(job.queue)
        if same_queue and ssn.queue_limit is not None and qi is not None:
            headroom = (ssn.queue_limit[qi] - ssn.queue_alloc[qi]) \
                .numpy().astype(np.float64).copy()

        placed = 0
        for tc in job.pending_classes():
            if placed >= still_needed:
                break
            rep0 = tc.tasks[0]
            if rep0.pod is not None and \
                    rep0.pod.preemption_policy == "Never":
                continue    # reference preempt.go:441 (#3642): a Never
                # preemptor may wait but must not displace anything
            req_vec = nt.req_vector(rep0)
            if req_vec is None:
                continue
            constraints = predicates.class_constraints(tc, job) if predicates \
                else (-1, np.zeros(max(nt.labels.words, 1), dtype=np.int64),
                      np.zeros(max(nt.labels.words, 1), dtype=np.int64))

            # K5 tensor pass: global victim filter once per
            # class, per-node evictable capacity as an extra-credit plane,
            # ONE fused score_cap over all nodes → ranked candidate nodes.
            # storm regime: thousands of identical
            # starving preemptors would each pay the full victim
            # collection + plugin intersection + scoring pass.  Identical
            # preemptor shapes (queue, priority, class signature) share
            # ONE scan per cycle; the walk itself stays exact because
            # _preempt_one revalidates fit and skips RELEASING victims.
            mkey = None
            if memo is not None:
                mkey = (job.queue, tc.tasks[0].priority, tc.signature)
                scan = memo.get(mkey, False)
                if scan is False:
                    scan = self._candidate_scan(
                        ssn, job, tc, req_vec, constraints, same_queue,
                        victim_filter)
                    memo[mkey] = scan
            else:
                scan = self._candidate_scan(ssn, job, tc, req_vec,
                                            constraints, same_queue,
                                            victim_filter)
            if scan is None:
                continue
            order, victims_by_node = scan
            for task in tc.tasks:
                if placed >= still_needed:
                    break
                node = self._preempt_one(ssn, stmt, job, task, req_vec,
                                         order, victims_by_node,
                                         headroom=headroom)
                if node is not None:
                    placed += 1
                    if headroom is not None:
                        headroom -= req_vec

        # commit on the JobPipelined VOTE alone (reference reclaim.go:163 /
        # preempt.go): with the gang plugin's pipelined gate enabled this
        # equals the occupied+waiting>=min check; with it disabled (conf
        # flags) partial progress commits — reference semantics
        if stmt.ops and ssn.job_pipelined(job):
            stmt.commit()
        else:
            stmt.discard()

    def _candidate_scan(self, ssn, job, tc, req_vec, constraints,
                        same_queue, victim_filter, node_filter=None):
        """One fused feasibility pass for the whole class: evictable
        resources per node become the kernel's future-credit plane.
        ``node_filter`` (a set of node names) restricts both victims and
        candidate nodes — the hypernode-domain trial path."""
        import torch
        from ...ops import reference as ref
        nt = ssn.node_tensors
        tol, require, forbid = constraints
        rep = tc.tasks[0]

        # global victim set (tier-intersected once, not per node)
        candidates = []
        for ni in ssn.nodes.values():
            if not ni.ready:
                continue
            if node_filter is not None and ni.name not in node_filter:
                continue
            for t in ni.tasks.values():
                if t.status not in self.victim_statuses:
                    continue
                if not t.preemptable:
                    continue   # explicit volcano.sh/preemptable=false
                vjob = ssn.jobs.get(t.job_key)
                if vjob is None:
                    continue
                if same_queue and vjob.queue != job.queue:
                    continue
                candidates.append(t)
        victims = victim_filter(rep, candidates) if victim_filter \
            else ssn.preemptable(rep, candidates)

        victims_by_node = {}
        evictable = np.zeros((nt.n, nt.r), dtype=np.float32)
        for v in victims:
            ni = ssn.nodes.get(v.node_name)
            if ni is None:
                continue
            victims_by_node.setdefault(ni.name, []).append(v)
            vv = nt.req_vector(v)
            if vv is not None:
                evictable[ni.node_id] += vv
        if not victims_by_node:
            return None
        for vs in victims_by_node.values():
            vs.sort(key=lambda v: victim_sort_key(ssn, v))

        dev = nt.alloc_t.device
        nt.ensure_plane_width()
        extra = nt.extra_t + torch.from_numpy(evictable.T).to(dev)
        W = nt.planes_t.shape[0]
        require = np.pad(require, (0, max(0, W - len(require))))
        forbid = np.pad(forbid, (0, max(0, W - len(forbid))))
        score = torch.empty(nt.n, device=dev)
        cap = torch.empty(nt.n, dtype=torch.int32, device=dev)
        ref.score_cap(nt.alloc_t.t(), nt.used_t.t(), extra.t(),
                      nt.ready.bool(), nt.taint_mask, nt.planes_t.t(),
                      torch.from_numpy(req_vec).to(dev), tol,
                      torch.from_numpy(require).to(dev),
                      torch.from_numpy(forbid).to(dev),
                      1.0, 0.0, 0.0, torch.ones(nt.r, device=dev), None,
                      score, cap)
        feasible = (score > float("-inf")).cpu().numpy()
        # stable: equal scores keep index order (CUDA argsort is otherwise
        # tie-unstable, which would pair tasks to nodes differently than
        # the CPU path — found by the topology-preempt equivalence test)
        order_idx = torch.argsort(score, descending=True,
                                  stable=True).cpu().numpy()
        nodes_sorted = getattr(ssn.cache, "nodes_sorted", None)
        if nodes_sorted is None or len(nodes_sorted) != len(ssn.nodes):
            nodes_sorted = sorted(ssn.nodes.values(), key=lambda n: n.name)
        order = [nodes_sorted[i] for i in order_idx if feasible[i]]
        if node_filter is not None:
            # domain trials must not pipeline OUTSIDE the domain even
            # when an out-of-domain node would fit without evictions
            order = [ni for ni in order if ni.name in node_filter]
        return order, victims_by_node

    def _preempt_one(self, ssn, stmt: Statement, job: JobInfo,
                     task: TaskInfo, req_vec, order, victims_by_node,
                     headroom=None) -> Optional[str]:
        """Place one preemptor task: walk the kernel-ranked candidate
        nodes, evict just-enough of the node's (pre-sorted) victims.
        ``headroom`` (same-queue quota slack) joins the per-node trial:
        a node whose victims cannot ALSO restore quota is skipped with
        no eviction committed — the reference's per-node nodeStmt
        isolation (preempt_test.go "only commit evictions on the node
        where preemption succeeds").  On success the freed quota is
        credited back into ``headroom`` in place."""
        nt = ssn.node_tensors
        didx = nt.dims.index

        def head_ok(gain):
            if headroom is None:
                return True
            for name, i in didx.items():
                if req_vec[i] > 0.1 and \
                        headroom[i] + gain.get(name, 0.0) + 0.1 < req_vec[i]:
                    return False
            return True

        for ni in order:
            no_gain = {}
            if _node_fits(nt, ni, req_vec) and head_ok(no_gain):
                stmt.pipeline(task, ni.name)
                return ni.name
            cands = victims_by_node.get(ni.name, [])
            # evict-just-enough simulation over future idle + quota
            chosen = []
            fi = ni.future_idle
            avail = {k: fi.get(k) for k in didx}
            gain = {}

            def fits():
                for name, i in didx.items():
                    if req_vec[i] > 0.1 and avail[name] + 0.1 < req_vec[i]:
                        return False
                return head_ok(gain)

            for v in cands:
                if fits():
                    break
                if v.status == TaskStatus.RELEASING:
                    continue           # already evicted for an earlier task
                chosen.append(v)
                for name, val in v.request.q.items():
                    if name in avail:
                        avail[name] += val
                    gain[name] = gain.get(name, 0.0) + val
            if not fits():
                continue               # per-node trial failed: no evictions
            for v in chosen:
                stmt.evict(v)
            if headroom is not None:
                for name, val in gain.items():
                    i = didx.get(name)
                    if i is not None:
                        headroom[i] += val
            stmt.pipeline(task, ni.name)
            return ni.name
        return None
