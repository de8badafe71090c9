"""Capacity plugin (reference ``plugins/capacity/capacity.go``, design
``docs/design/capacity-scheduling.md``) — the modern replacement for
proportion: explicit per-queue ``deserved`` from the Queue spec, elastic
capacity ``guarantee → deserved → capability``, and **hierarchical
queues** (children split their parent's deserved share by weight;
ancestor capability caps the subtree — buildHierarchicalQueueAttrs
capacity.go:1212, ancestor checks :1452-1560).

Tensorized: the hierarchy is resolved level by level with the same
float64 water-filling kernel as proportion (ops.reference.waterfill) —
one pass per tree level instead of per-queue Go recursion.
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np
import torch

from ...api.types import PodGroupPhase
from ...ops import reference as ref
from ..plan import BIG_LIMIT
from ..session import PERMIT, REJECT
from .base import Plugin, register


@register("capacity")
class CapacityPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        queues = [ssn.queues[name] for name in sorted(ssn.queues)]
        Q, R = len(queues), nt.r
        if Q == 0:
            return
        total = ssn.total_resource if ssn.total_resource is not None else \
            torch.zeros(R)
        by_name = {q.name: q for q in queues}
        qi_of = ssn.queue_index

        # request per queue (demand) — f64 accumulation (see proportion.py)
        req_np = np.zeros((Q, R), dtype=np.float64)
        for job in ssn.jobs.values():
            qi = qi_of.get(job.queue)
            if qi is not None:
                req_np[qi] += job.total_req_vec(nt)
        request = torch.from_numpy(req_np).to(torch.float32)

        # per-queue vectors
        guarantee = torch.zeros((Q, R), dtype=torch.float32)
        capability = torch.full((Q, R), BIG_LIMIT, dtype=torch.float32)
        deserved_spec = torch.zeros((Q, R), dtype=torch.float32)
        has_deserved = [False] * Q
        for q in queues:
            i = qi_of[q.name]
            guarantee[i] = torch.from_numpy(nt.resource_vector(q.guarantee))
            cv = nt.resource_vector(q.capability)
            for r in range(R):
                if cv[r] > 0:
                    capability[i, r] = float(cv[r])
            dv = nt.resource_vector(q.deserved_spec)
            if dv.any():
                deserved_spec[i] = torch.from_numpy(dv)
                has_deserved[i] = True

        # hierarchy: children of each parent ("" = root level)
        children: Dict[str, List[str]] = {}
        for q in queues:
            parent = q.parent if q.parent in by_name else ""
            children.setdefault(parent, []).append(q.name)

        # aggregate subtree demand bottom-up so parents see child demand
        def subtree_request(name: str) -> torch.Tensor:
            req = request[qi_of[name]].clone()
            for c in children.get(name, []):
                req += subtree_request(c)
            return req

        deserved = torch.zeros((Q, R), dtype=torch.float32)

        # level-order: split each parent's pool among its children.
        # Two passes: explicit spec.deserved dims are clamped first; the
        # remaining pool water-fills by weight among the rest (so slack
        # from a low explicit deserved flows to siblings).
        def fill_level(parent: str, pool: torch.Tensor) -> None:
            names = children.get(parent, [])
            if not names:
                return
            idx = [qi_of[n] for n in names]
            cap = {n: torch.minimum(capability[qi_of[n]], pool)
                   for n in names}
            fixed = torch.zeros_like(pool)
            for n in names:
                i = qi_of[n]
                if has_deserved[i]:
                    spec = torch.minimum(deserved_spec[i], cap[n])
                    spec = torch.maximum(spec, guarantee[i])
                    fixed = fixed + torch.where(deserved_spec[i] > 0, spec,
                                                torch.zeros_like(spec))
            rest_pool = torch.clamp(pool - fixed, min=0.0)
            w = torch.tensor([by_name[n].weight for n in names],
                             dtype=torch.float32)
            req = torch.stack([subtree_request(n) for n in names])
            # explicit-deserved dims don't compete in the water-fill
            for k, n in enumerate(names):
                i = qi_of[n]
                if has_deserved[i]:
                    req[k] = torch.where(deserved_spec[i] > 0,
                                         torch.zeros_like(req[k]), req[k])
            gua = guarantee[idx]
            cap_m = torch.stack([cap[n] for n in names])
            des = ref.waterfill(w, req, gua, cap_m, rest_pool)
            for k, n in enumerate(names):
                i = qi_of[n]
                if has_deserved[i]:
                    spec_k = torch.minimum(deserved_spec[i], cap[n])
                    spec_k = torch.maximum(spec_k, guarantee[i])
                    deserved[i] = torch.where(deserved_spec[i] > 0, spec_k,
                                              des[k])
                else:
                    deserved[i] = des[k]
                fill_level(n, deserved[i])

        fill_level("", total.clone())
        ssn.queue_deserved = deserved
        limit = torch.where(total.unsqueeze(0) > 0, deserved,
                            torch.full_like(deserved, BIG_LIMIT))
        ssn.queue_limit = torch.maximum(limit, ssn.queue_alloc)
        # hierarchical-DRF equilibrium caps (plugins/drf.py) must survive
        # this row replacement regardless of plugin tier order
        for _qi, _cap in getattr(ssn, "hdrf_cap_rows", {}).items():
            ssn.queue_limit[_qi] = torch.minimum(ssn.queue_limit[_qi], _cap)
        self.deserved = deserved

        alloc_np = ssn.queue_alloc.numpy()
        des_np = deserved.numpy()
        total_np = total.numpy()
        mask = total_np > 0
        des64 = des_np.astype(np.float64)

        def q_share(qi: int) -> float:
            # allocated/deserved per dim (reference capacity.go:1812-1824;
            # no-deserved best-effort queues pin to share 1)
            d = des64[qi][mask]
            pos = d > 0
            a = alloc_np[qi][mask].astype(np.float64)
            if not pos.any():
                return 1.0
            s = float((a[pos] / d[pos]).max())
            if (a[~pos] > 0.1).any():
                s = max(s, 1.0)
            return s

        def queue_order(a, b) -> int:
            # priority first (capacity honors queue priority), then share
            if a.priority != b.priority:
                return -1 if a.priority > b.priority else 1
            sa, sb = q_share(qi_of[a.name]), q_share(qi_of[b.name])
            if abs(sa - sb) < 1e-12:
                return 0
            return -1 if sa < sb else 1

        def overused(q) -> bool:
            qi = qi_of[q.name]
            return bool((alloc_np[qi][mask] > des_np[qi][mask] + 0.1).any())

        def allocatable(q, job) -> bool:
            return not overused(q)

        inqueue_np = np.zeros((Q, R), dtype=np.float64)
        for job in ssn.jobs.values():
            if job.phase == PodGroupPhase.INQUEUE.value:
                qi = qi_of.get(job.queue)
                if qi is not None:
                    inqueue_np[qi] += job.minres_vec(nt)

        def job_enqueueable(job) -> int:
            qi = qi_of.get(job.queue)
            if qi is None:
                return REJECT
            minres = job.minres_vec(nt)
            if (minres[~mask] > 0.1).any():
                return REJECT     # demand on a zero-capacity dim
            head = (alloc_np[qi][mask].astype(np.float64)
                    + inqueue_np[qi][mask] + minres[mask])
            des = des_np[qi][mask].astype(np.float64)
            return PERMIT if bool((head <= des + 0.1 + 1e-6 * des).all()) \
                else REJECT

        def job_enqueued(job) -> None:
            qi = qi_of.get(job.queue)
            if qi is not None:
                inqueue_np[qi] += job.minres_vec(nt)

        # -- reclaim eligibility with ancestor levels ------------------------
        # (reference capacity.go:500-600 AddReclaimableFn + the level
        # semantics of capacity_test.go cases 1-11)
        anc_level = int(self.args.get("ancestorReclaimLevel", 0) or 0)
        des_spec_np = deserved_spec.numpy().astype(np.float64)
        gua_np = guarantee.numpy().astype(np.float64)
        # ancestors chain per queue, root-first, excluding self
        # ("" = virtual root, never counted)
        anc_chain: Dict[str, List[str]] = {}
        for q in queues:
            chain, cur, seen = [], q.queue.spec.parent, set()
            while cur and cur in by_name and cur not in seen:
                seen.add(cur)
                chain.append(cur)
                cur = by_name[cur].queue.spec.parent
            anc_chain[q.name] = chain[::-1]       # root → parent

        desc_rows: Dict[str, List[int]] = {q.name: [qi_of[q.name]]
                                           for q in queues}
        for q in queues:
            for a in anc_chain[q.name]:
                desc_rows.setdefault(a, []).append(qi_of[q.name])

        def _is_root(qn: str) -> bool:
            q = by_name.get(qn)
            return q is None or not q.queue.spec.parent

        def _anc_at(qname: str, level: int):
            chain = anc_chain.get(qname, ())
            return chain[len(chain) - level] if 0 < level <= len(chain) \
                else None

        def _shares_nonroot_anc(a: str, b: str) -> bool:
            aa = {x for lv in range(1, anc_level + 1)
                  if (x := _anc_at(a, lv)) is not None and not _is_root(x)}
            bb = {x for lv in range(1, anc_level + 1)
                  if (x := _anc_at(b, lv)) is not None and not _is_root(x)}
            return bool(aa & bb)

        def reclaimable(reclaimer, candidates):
            rjob = ssn.jobs.get(reclaimer.job_key)
            rq = rjob.queue if rjob is not None else ""
            rreq = nt.req_vector(reclaimer)
            rreq = rreq.astype(np.float64) if rreq is not None else None
            rdes = des_spec_np[qi_of[rq]] if rq in qi_of else None
            allocations: Dict[str, np.ndarray] = {}

            def alloc_of(qn: str) -> np.ndarray:
                a = allocations.get(qn)
                if a is None:
                    rows = desc_rows.get(qn, [])
                    a = alloc_np[rows].astype(np.float64).sum(axis=0)
                    allocations[qn] = a
                return a

            out = []
            for v in candidates:
                job = ssn.jobs.get(v.job_key)
                if job is None:
                    continue
                q = ssn.queues.get(job.queue)
                if q is None or not q.reclaimable or job.queue not in qi_of:
                    continue
                vreq = nt.req_vector(v)
                if vreq is None:
                    continue
                vreq = vreq.astype(np.float64)
                # no intersecting dims with the reclaimer → irrelevant
                if rreq is not None and not ((vreq > 0) & (rreq > 0)).any():
                    continue
                qi = qi_of[job.queue]
                alloc = alloc_of(job.queue)
                # guarantee floor: evicting must not dip below guarantee
                if ((alloc - vreq) < gua_np[qi] - 1e-9).any():
                    continue
                des = des_spec_np[qi]
                relevant = (vreq > 0) & (des > 0)
                if not relevant.any():
                    # immediate victim (leaf deserved silent on its dims)
                    if anc_level > 0 and rq and \
                            _shares_nonroot_anc(rq, job.queue) and \
                            not (rreq is not None and rdes is not None
                                 and ((rreq > 0) & (rdes > 0)).any()):
                        continue    # no real contention under shared scope
                elif not (alloc[relevant] > des[relevant] + 0.1).any():
                    continue        # within deserved on every relevant dim
                # ancestor gates up to the configured level
                ok = True
                touched = []
                for lv in range(1, anc_level + 1):
                    anc = _anc_at(job.queue, lv)
                    if anc is None or _is_root(anc) \
                            or anc == _anc_at(rq, lv):
                        continue    # reference skips the tree root and
                        # shared ancestors (getReclaimeeAncestorToCheck)
                    a_alloc = alloc_of(anc)
                    a_des = des_spec_np[qi_of[anc]] if anc in qi_of else None
                    if a_des is None:
                        ok = False
                        break
                    rel = (vreq > 0) & (a_des > 0)
                    if not rel.any():
                        touched.append(a_alloc)
                        continue    # immediate victim vs this ancestor
                    if not (a_alloc[rel] > a_des[rel] + 0.1).any():
                        ok = False
                        break
                    touched.append(a_alloc)
                if not ok:
                    continue
                alloc -= vreq
                for a in touched:
                    a -= vreq
                out.append(v)
            return out

        ssn.add_queue_order_fn(
            queue_order,
            key=lambda q: (-q.priority, q_share(qi_of[q.name])))
        def job_enqueueable_bulk(qname, jobs, rows=None, table=None):
            qi = qi_of.get(qname)
            if qi is None:
                return None
            if rows is not None and table is not None \
                    and table.minres.shape[1] == R:
                demand = table.minres[rows].sum(axis=0)
            else:
                demand = np.zeros(R, dtype=np.float64)
                for j in jobs:
                    demand += j.minres_vec(nt)
            if (demand[~mask] > 0.1).any():
                return None       # demand on a zero-capacity dim
            head = (alloc_np[qi][mask].astype(np.float64)
                    + inqueue_np[qi][mask] + demand[mask])
            des = des_np[qi][mask].astype(np.float64)
            if bool((head <= des + 0.1 + 1e-6 * des).all()):
                return lambda: inqueue_np[qi].__iadd__(demand)
            return None

        ssn.overused_fns.append(overused)
        ssn.allocatable_fns.append(allocatable)
        ssn.job_enqueueable_fns.append(job_enqueueable)
        ssn.job_enqueueable_bulk_fns.append(job_enqueueable_bulk)
        ssn.job_enqueued_fns = getattr(ssn, "job_enqueued_fns", [])
        ssn.job_enqueued_fns.append(job_enqueued)
        ssn.reclaimable_fns.append(reclaimable)
