"""Proportion plugin (reference ``plugins/proportion/proportion.go``).

Queue fair share: iterative weighted water-filling of ``deserved``
capped by capability and demand (proportion.go:90-260), computed as one
tensor pass (ops.reference.waterfill, float64 accumulators for
reproducibility).  Registers: QueueOrder by share (:268), Overused
(:321), Allocatable → the per-queue ``queue_limit`` row the select/commit
kernel enforces ON DEVICE (:359 — in the reference this is a host
callback per allocation; here the quota is a tensor bound the kernel
clamps against, so enforcement costs nothing per task), JobEnqueueable
(:404), Reclaimable (:288).

Host-side checks run on zero-copy numpy views of the queue tensors —
the per-job vote is a handful of numpy scalar ops (hot: enqueue at 10k+
jobs/cycle).
"""

from __future__ import annotations

import numpy as np
import torch

from ...ops import reference as ref
from ..plan import BIG_LIMIT
from ..session import PERMIT, REJECT
from .base import Plugin, register


@register("proportion")
class ProportionPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        queues = [ssn.queues[name] for name in sorted(ssn.queues)]
        Q, R = len(queues), nt.r
        if Q == 0:
            return
        total = ssn.total_resource if ssn.total_resource is not None else \
            torch.zeros(R)

        weight = torch.tensor([q.weight for q in queues], dtype=torch.float32)
        # f64 accumulation: at 1M-pod scale the per-queue demand sum sits
        # at ~1e9 where f32 ulp is 64 — f32 accumulation under-counts by
        # enough to starve the tail gang's quota
        req_np = np.zeros((Q, R), dtype=np.float64)
        guarantee = torch.zeros((Q, R), dtype=torch.float32)
        capability = torch.full((Q, R), BIG_LIMIT, dtype=torch.float32)
        for i, q in enumerate(queues):
            guarantee[i] = torch.from_numpy(nt.resource_vector(q.guarantee))
            cap_vec = nt.resource_vector(q.capability)
            for r in range(R):
                if cap_vec[r] > 0:
                    capability[i, r] = float(cap_vec[r])
        jt = getattr(ssn, "job_table", None)
        if jt is None or len(jt.jobs) != len(ssn.jobs) \
                or jt.totreq.shape[1] != R:
            jt = None
        if jt is not None:
            ok = jt.qi >= 0
            np.add.at(req_np, jt.qi[ok], jt.totreq[ok])
        else:
            for job in ssn.jobs.values():
                qi = ssn.queue_index.get(job.queue)
                if qi is not None:
                    req_np[qi] += job.total_req_vec(nt)
        request = torch.from_numpy(req_np).to(torch.float32)

        deserved = ref.waterfill(weight, request, guarantee, capability, total)
        ssn.queue_deserved = deserved
        # Allocatable enforcement: the kernel clamps each placement against
        # queue_limit - queue_alloc; deserved IS the limit (proportion
        # semantics: a queue may not allocate past its deserved share).
        # Dims with zero cluster total stay unlimited (count-less dims).
        limit = torch.where(total.unsqueeze(0) > 0, deserved,
                            torch.full_like(deserved, BIG_LIMIT))
        ssn.queue_limit = torch.maximum(limit, ssn.queue_alloc)
        # hierarchical-DRF equilibrium caps (plugins/drf.py) must survive
        # this row replacement regardless of plugin tier order
        for _qi, _cap in getattr(ssn, "hdrf_cap_rows", {}).items():
            ssn.queue_limit[_qi] = torch.minimum(ssn.queue_limit[_qi], _cap)
        self.deserved = deserved

        # zero-copy numpy views for the hot host-side checks
        alloc_np = ssn.queue_alloc.numpy()
        des_np = deserved.numpy()
        total_np = total.numpy()
        mask = total_np > 0

        des64 = des_np.astype(np.float64)

        def q_share(qi: int) -> float:
            # share_r = allocated_r / deserved_r (reference proportion.go
            # updateQueueAttrShare); deserved==0 dims: share 1 if anything
            # is allocated there, and a queue with NO deserved at all is
            # share 1 (best-effort sorts last, capacity.go:1820)
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
            sa = q_share(ssn.queue_index[a.name])
            sb = q_share(ssn.queue_index[b.name])
            if abs(sa - sb) < 1e-12:
                return 0
            return -1 if sa < sb else 1

        def overused(q) -> bool:
            qi = ssn.queue_index[q.name]
            return bool((alloc_np[qi][mask] > des_np[qi][mask] + 0.1).any())

        def allocatable(q, job) -> bool:
            # coarse host-side gate; the exact bound is enforced in-kernel
            return not overused(q)

        # inqueue accounting (proportion.go:404-441: enqueueable checks
        # allocated + already-admitted-but-unscheduled against deserved)
        from ...api.types import PodGroupPhase
        inqueue_np = np.zeros((Q, R), dtype=np.float64)
        if jt is not None:
            from ..jobtable import PH_INQUEUE
            sel = (jt.qi >= 0) & (jt.phase == PH_INQUEUE)
            np.add.at(inqueue_np, jt.qi[sel], jt.minres[sel])
        else:
            for job in ssn.jobs.values():
                if job.phase == PodGroupPhase.INQUEUE.value:
                    qi = ssn.queue_index.get(job.queue)
                    if qi is not None:
                        inqueue_np[qi] += job.minres_vec(nt)

        def job_enqueueable(job) -> int:
            qi = ssn.queue_index.get(job.queue)
            if qi is None:
                return REJECT
            minres = job.minres_vec(nt)
            if (minres[~mask] > 0.1).any():
                return REJECT     # demand on a zero-capacity dim
            head = (alloc_np[qi][mask].astype(np.float64)
                    + inqueue_np[qi][mask] + minres[mask])
            des = des_np[qi][mask].astype(np.float64)
            # slack: absolute MIN_RESOURCE + f32 accumulation tolerance
            return PERMIT if bool((head <= des + 0.1 + 1e-6 * des).all()) \
                else REJECT

        def job_enqueued(job) -> None:
            qi = ssn.queue_index.get(job.queue)
            if qi is not None:
                inqueue_np[qi] += job.minres_vec(nt)

        def reclaimable(reclaimer, candidates):
            # a victim whose queue is over its deserved share may be
            # reclaimed (proportion.go:288), if the queue allows it
            out = []
            for v in candidates:
                job = ssn.jobs.get(v.job_key)
                if job is None:
                    continue
                q = ssn.queues.get(job.queue)
                if q is None or not q.reclaimable:
                    continue
                if overused(q):
                    out.append(v)
            return out

        def job_enqueueable_bulk(qname, jobs, rows=None, table=None):
            """Whole-queue batch vote: total pending demand under deserved
            <=> every sequential per-job vote permits (monotone inqueue
            sum).  Thunk commits the accounting once all plugins agree.
            With JobTable rows the demand sum is one vectorized reduce."""
            qi = ssn.queue_index.get(qname)
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

        ssn.add_queue_order_fn(
            queue_order, key=lambda q: q_share(ssn.queue_index[q.name]))
        ssn.overused_fns.append(overused)
        ssn.allocatable_fns.append(allocatable)
        ssn.job_enqueueable_fns.append(job_enqueueable)
        ssn.job_enqueueable_bulk_fns.append(job_enqueueable_bulk)
        ssn.job_enqueued_fns = getattr(ssn, "job_enqueued_fns", [])
        ssn.job_enqueued_fns.append(job_enqueued)
        ssn.reclaimable_fns.append(reclaimable)
