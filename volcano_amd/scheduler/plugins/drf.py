"""DRF plugin (reference ``plugins/drf/drf.go``).

Dominant share per job — ``share = max_r(allocated_r / total_r)``
(drf.go calculateShare) — computed as ONE vectorized pass over the [J, R]
allocation matrix (float64), not per-job Go loops.  JobOrder: lower share
first (:388).  Preemptable: victim's job share must stay above the
preemptor's (:263).

Note on the reference's share-update event handlers (drf.go:391): the
reference mutates shares *during* the cycle because its allocation is
incremental per task.  In the plan design the job order is frozen when
the cycle plan is built, so mid-cycle share updates cannot influence any
decision — shares are recomputed at the next session open instead.
"""

from __future__ import annotations

from typing import Dict, Optional

import numpy as np

from .base import Plugin, register

SHARE_DELTA = 0.000001      # reference drf.go shareDelta


class _HNode:
    """One node of the hierarchical fair-share tree (reference
    drf.go hierarchicalNode).  Leaves are jobs; internal nodes are the
    slash-path segments of the queues' ``volcano.sh/hierarchy``."""

    __slots__ = ("name", "weight", "children", "alloc", "request",
                 "share", "saturated")

    def __init__(self, name: str, weight: float, r: int):
        self.name = name
        self.weight = max(weight, 1.0)
        self.children: Optional[Dict[str, "_HNode"]] = {}
        self.alloc = np.zeros(r, dtype=np.float64)
        self.request = np.zeros(r, dtype=np.float64)
        self.share = 0.0
        self.saturated = False


class HdrfTree:
    """Weighted hierarchical DRF (reference drf.go:455-560
    buildHierarchy/updateHierarchicalShare).  The MI355X reformulation
    solves the reference's per-pod incremental allocation loop as a
    host-side *progressive-filling simulation* over fluid per-queue
    demand, producing per-queue caps the in-kernel queue clamp then
    enforces — one kernel cycle instead of 100k share updates."""

    def __init__(self, total: np.ndarray):
        self.total = np.maximum(total.astype(np.float64), 0.0)
        self.r = len(total)
        self.root = _HNode("root", 1.0, self.r)
        self.total_allocated = np.zeros(self.r, dtype=np.float64)
        self.queue_leafpath: Dict[str, list] = {}

    def _share(self, alloc: np.ndarray) -> float:
        m = self.total > 0
        if not m.any():
            return 0.0
        return float((alloc[m] / self.total[m]).max()) if len(alloc) else 0.0

    def ensure_queue(self, qname: str, hierarchy: str, weights: str) -> list:
        """Build the path nodes for a queue; returns [root..leaf-parent]."""
        path = self.queue_leafpath.get(qname)
        if path is not None:
            return path
        parts = (hierarchy or f"root/{qname}").split("/")
        wparts = (weights or "").split("/")
        node = self.root
        path = [node]
        for i in range(1, len(parts)):
            child = node.children.get(parts[i])
            if child is None:
                try:
                    w = float(wparts[i]) if i < len(wparts) else 1.0
                except ValueError:
                    w = 1.0
                child = _HNode(parts[i], w, self.r)
                node.children[parts[i]] = child
            node = child
            path.append(node)
        self.queue_leafpath[qname] = path
        return path

    def add_job(self, key: str, qname: str, hierarchy: str, weights: str,
                alloc: np.ndarray, request: np.ndarray) -> _HNode:
        parent = self.ensure_queue(qname, hierarchy, weights)[-1]
        leaf = _HNode(key, 1.0, self.r)
        leaf.children = None
        leaf.alloc = alloc.astype(np.float64).copy()
        leaf.request = request.astype(np.float64)
        parent.children[key] = leaf
        self.total_allocated += leaf.alloc
        return leaf

    def update(self) -> None:
        """Recompute shares/saturation bottom-up (drf.go:503-548)."""
        demanding = self.total_allocated < self.total
        self._update(self.root, demanding)

    def _update(self, node: _HNode, demanding: np.ndarray) -> None:
        if node.children is None:       # job leaf (resourceSaturated)
            a, q = node.alloc, node.request
            node.saturated = bool(
                ((a != 0) & (q != 0) & (a >= q)).any()
                or ((~demanding) & (q != 0)).any())
            node.share = self._share(a)
            return
        mdr = 1.0
        for c in node.children.values():
            self._update(c, demanding)
            if c.share != 0 and not c.saturated:
                s = self._share(c.alloc)
                if s < mdr:
                    mdr = s
        alloc = np.zeros(self.r, dtype=np.float64)
        saturated = True
        for c in node.children.values():
            if not c.saturated:
                saturated = False
            if c.share != 0:
                alloc += c.alloc if c.saturated \
                    else c.alloc * (mdr / c.share)
        node.alloc = alloc
        node.share = self._share(alloc)
        node.saturated = saturated

    def compare_queues(self, lq: str, rq: str) -> float:
        """Walk the two paths level by level (drf.go:160-186): saturated
        nodes lose; otherwise weighted share decides; ties descend."""
        lp = self.queue_leafpath.get(lq)
        rp = self.queue_leafpath.get(rq)
        if lp is None or rp is None:
            return 0.0
        for ln, rn in zip(lp, rp):
            if not ln.saturated and rn.saturated:
                return -1.0
            if ln.saturated and not rn.saturated:
                return 1.0
            ls, rs = ln.share / ln.weight, rn.share / rn.weight
            if ls != rs:
                return ls - rs
        return 0.0


@register("drf")
class DrfPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        jt = getattr(ssn, "job_table", None)
        if jt is not None and len(jt.jobs) == len(ssn.jobs):
            jobs = jt.jobs
            occ_rows = np.nonzero(jt.occ > 0)[0]
        else:
            jobs = list(ssn.jobs.values())
            occ_rows = np.array([i for i, j in enumerate(jobs)
                                 if j.occupied_count], dtype=np.int64)
        if not len(jobs):
            return
        # zero-alloc jobs (the common start-of-cycle state) skip the
        # vector build entirely — share is exactly 0
        if not len(occ_rows):
            self.share = {}
        else:
            alloc = np.zeros((len(jobs), nt.r), dtype=np.float64)
            for i in occ_rows:
                alloc[i] = jobs[i].alloc_vec(nt)
            total = ssn.total_resource.numpy().astype(np.float64) \
                if ssn.total_resource is not None else np.ones(nt.r)
            shares = (alloc / np.maximum(total, 1.0)).max(axis=1)     # [J]
            self.share = {job.key: float(shares[i])
                          for i, job in enumerate(jobs)}

        def job_order(a, b) -> int:
            sa = self.share.get(a.key, 0.0)
            sb = self.share.get(b.key, 0.0)
            if abs(sa - sb) < 1e-12:
                return 0
            return -1 if sa < sb else 1

        def preemptable(preemptor, candidates):
            ps = self.share.get(preemptor.job_key, 0.0)
            return [v for v in candidates
                    if self.share.get(v.job_key, 0.0) > ps]

        any_share = bool(self.share) and any(
            v != 0.0 for v in self.share.values())

        def share_col(jt, rows):
            if not any_share:
                return np.zeros(len(rows))
            sh = self.share
            return np.fromiter((sh.get(k, 0.0) for k in jt.keys[rows]),
                               dtype=np.float64, count=len(rows))

        ssn.add_job_order_fn(
            job_order, key=lambda j: self.share.get(j.key, 0.0),
            col=share_col)
        ssn.preemptable_fns.append(preemptable)

        if self.args.get("enableHierarchy"):
            self._open_hierarchy(ssn, nt)

    # -- hierarchical DRF (reference drf.go EnabledHierarchy) ----------------
    def _open_hierarchy(self, ssn, nt) -> None:
        """Build the weighted share tree, solve the HDRF equilibrium by
        progressive filling over fluid per-queue demand, and publish the
        result as per-queue allocation caps for the in-kernel clamp.
        The reference reaches the same fixed point by re-sorting queues
        after every single pod placement (drf.go:265-369 + hdrf_test.go
        "rescaling": complementary demands each get 50% of their
        dominant resource) — here the fixed point is computed once per
        cycle on the host, which is equivalent in the fluid limit and
        keeps the device path single-pass."""
        import torch
        from ...api.types import PodGroupPhase, TaskStatus
        total = ssn.total_resource.numpy().astype(np.float64) \
            if ssn.total_resource is not None else np.zeros(nt.r)
        tree = HdrfTree(total)
        self.htree = tree

        # per-job leaves: current allocation + total request; pending
        # demand in single-task quanta for the filling simulation
        pend = []        # (leaf, qname, quantum_vec, count)
        active = (PodGroupPhase.INQUEUE.value, PodGroupPhase.RUNNING.value)
        for job in ssn.jobs.values():
            q = ssn.queues.get(job.queue)
            if q is None:
                continue
            leaf = tree.add_job(job.key, q.name, q.hierarchy,
                                q.hierarchy_weights,
                                job.alloc_vec(nt), job.total_req_vec(nt))
            npend = len(job.task_status_index.get(TaskStatus.PENDING, ()))
            if npend and job.phase in active and len(job.tasks):
                quantum = job.total_req_vec(nt) / max(len(job.tasks), 1)
                pend.append([leaf, q.name, quantum.astype(np.float64),
                             npend])
        tree.update()

        def queue_order(a, b) -> int:
            ret = tree.compare_queues(a.name, b.name)
            return -1 if ret < 0 else (1 if ret > 0 else 0)

        def queue_key(q):
            path = tree.queue_leafpath.get(q.name)
            if not path:
                return ()
            # same order as compare_queues: saturated loses the level,
            # then weighted share, ties descend (lexicographic tuple)
            return tuple((n.saturated, n.share / n.weight) for n in path)

        ssn.add_queue_order_fn(queue_order, key=queue_key)

        def reclaimable(reclaimer, candidates):
            """Hierarchical victim filter (drf.go:281-366): evicting the
            candidate must leave the reclaimer's queue strictly poorer
            than the victim's in the weighted tree order."""
            ljob = ssn.jobs.get(reclaimer.job_key)
            if ljob is None:
                return candidates
            lq = ljob.queue
            lleaf = self._leaf(tree, ljob.key)
            req = nt.req_vector(reclaimer)
            if lleaf is None or req is None:
                return candidates
            victims = []
            req = req.astype(np.float64)
            lleaf.alloc += req
            tree.total_allocated += req
            for v in candidates:
                rjob = ssn.jobs.get(v.job_key)
                vleaf = rjob and self._leaf(tree, rjob.key)
                vreq = nt.req_vector(v)
                if vleaf is None or vreq is None:
                    continue
                vreq = vreq.astype(np.float64)
                vleaf.alloc -= vreq
                tree.total_allocated -= vreq
                tree.update()
                if tree.compare_queues(lq, rjob.queue) < 0:
                    victims.append(v)
                vleaf.alloc += vreq
                tree.total_allocated += vreq
            lleaf.alloc -= req
            tree.total_allocated -= req
            tree.update()
            return victims

        ssn.reclaimable_fns.append(reclaimable)

        if not pend:
            return
        # progressive filling: repeatedly give one task quantum to the
        # poorest unsaturated queue until the cluster (or demand) is dry
        caps = {}        # qname -> np vector of granted capacity
        free = np.maximum(total - tree.total_allocated, 0.0)
        guard = sum(p[3] for p in pend) + 1
        while guard > 0:
            guard -= 1
            tree.update()
            best = None
            for p in pend:
                if p[3] <= 0:
                    continue
                need = p[2]
                # relative tolerance: f32-sourced byte counts drift at
                # the 1e-7 scale, absolute epsilons don't cover 1e9-byte
                # dims
                if ((need > free + 1e-9 + 1e-6 * need) & (need > 0)).any():
                    continue
                if best is None:
                    best = p
                    continue
                ret = tree.compare_queues(p[1], best[1])
                # same-queue entries order by leaf (job) dominant share —
                # the reference's within-queue DRF job order, so
                # complementary-demand jobs in one queue alternate
                if ret < 0 or (ret == 0
                               and p[0].share < best[0].share - 1e-12):
                    best = p
            if best is None:
                break
            leaf, qn, quantum, _ = best
            leaf.alloc += quantum
            tree.total_allocated += quantum
            free -= quantum
            best[3] -= 1
            prev = caps.get(qn)
            caps[qn] = quantum.copy() if prev is None else prev + quantum
        if caps and ssn.queue_limit is not None and ssn.queue_index:
            lim = ssn.queue_limit
            hdrf_rows = {}
            for qn, extra in caps.items():
                qi = ssn.queue_index.get(qn)
                if qi is None:
                    continue
                base = ssn.queue_alloc[qi] if ssn.queue_alloc is not None \
                    else torch.zeros(nt.r)
                # relative slack ≫ f32 rounding of big byte counts but
                # ≪ one task quantum, so the clamp never eats a slot
                hdrf_cap = (base + torch.from_numpy(
                    extra.astype(np.float32))) * (1.0 + 1e-5)
                lim[qi] = torch.minimum(lim[qi], hdrf_cap)
                hdrf_rows[qi] = hdrf_cap
            # proportion/capacity REPLACE ssn.queue_limit when they open
            # after drf in the tier order — publish the equilibrium caps
            # so they re-apply them (queue-limit row combination)
            ssn.hdrf_cap_rows = hdrf_rows

    @staticmethod
    def _leaf(tree: HdrfTree, job_key: str):
        for path in tree.queue_leafpath.values():
            leaf = path[-1].children.get(job_key)
            if leaf is not None:
                return leaf
        return None
