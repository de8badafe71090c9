"""Per-cycle Session: snapshot + plugin callback registries.

Analog of the reference's ``pkg/scheduler/framework/session.go`` (the
Session struct with ~35 extension-point registries and the tiered
aggregation in ``session_plugins.go``), re-shaped for the tensor decision
plane:

* callbacks that *order* work (queue/job/task order) or *vote* (enqueueable,
  pipelined) stay host-side functions — they run over Q/J-sized data;
* callbacks that in the reference run per (task, node) — predicates and
  node scores — become **tensor configuration**: plugins contribute label
  bit-planes, taint masks, score weights and queue limit rows that the HIP
  kernels consume.

Tier semantics (reference framework/session_plugins.go): order functions
are compared tier-by-tier, first non-zero wins; JobReady is AND over all
registered fns (:483); enqueueable/pipelined are votes — any reject wins,
else any permit, else default (:582/:505); victim functions intersect
per tier (:225/:274).
"""

from __future__ import annotations

import time
from typing import Callable, Dict, List, Optional

import torch

from ..api.info import JobInfo, NodeInfo, QueueInfo, TaskInfo
from .plan import BIG_LIMIT
from .tensors import NodeTensors

# vote values (reference util.Permit/Abstain/Reject)
PERMIT, ABSTAIN, REJECT = 1, 0, -1

CmpFn = Callable[[object, object], int]


class Session:
    def __init__(self, cache, config):
        self.cache = cache
        self.config = config
        self.uid = f"ssn-{time.time_ns()}"

        # snapshot views (built by cache.snapshot_into)
        self.jobs: Dict[str, JobInfo] = {}
        self.nodes: Dict[str, NodeInfo] = {}
        self.queues: Dict[str, QueueInfo] = {}
        self.node_tensors: Optional[NodeTensors] = None
        self.queue_index: Dict[str, int] = {}
        self.total_resource: Optional[torch.Tensor] = None  # [R] f32

        # host-side registries, keyed per tier: List[List[fn]]
        self.job_order_fns: List[List[CmpFn]] = []
        self.queue_order_fns: List[List[CmpFn]] = []
        self.task_order_fns: List[List[CmpFn]] = []
        self.job_order_keys: List[List] = []
        self.job_order_cols: List[List] = []   # vectorized key providers
        self.queue_order_keys: List[List] = []
        self.job_valid_fns: List[Callable[[JobInfo], bool]] = []
        # vectorized job_valid counterparts: fn(job_table) -> bool mask [J].
        # The columnar worksheet path applies them only when EVERY
        # job_valid fn has one (1:1), else it falls back per job.
        self.job_valid_cols: List[Callable] = []
        self.job_ready_fns: List[Callable[[JobInfo], bool]] = []
        self.job_pipelined_fns: List[Callable[[JobInfo], int]] = []
        self.job_enqueueable_fns: List[Callable[[JobInfo], int]] = []
        # optional bulk counterparts: fn(queue_name, jobs) -> True when the
        # WHOLE batch can be admitted (and internal accounting updated),
        # None to fall back to per-job votes.  Exact when admission
        # depends on monotone accumulated sums (proportion/overcommit).
        self.job_enqueueable_bulk_fns: List[Callable] = []
        self.job_starving_fns: List[Callable[[JobInfo], bool]] = []
        self.overused_fns: List[Callable[[QueueInfo], bool]] = []
        self.allocatable_fns: List[Callable[[QueueInfo, JobInfo], bool]] = []
        self.preemptable_fns: List[
            Callable[[TaskInfo, List[TaskInfo]], List[TaskInfo]]] = []
        self.reclaimable_fns: List[
            Callable[[TaskInfo, List[TaskInfo]], List[TaskInfo]]] = []
        self.victim_tasks_fns: List[
            Callable[[List[TaskInfo]], List[TaskInfo]]] = []
        # filters applied over the nominated victim union (pdb/conformance)
        self.victim_filter_fns: List[
            Callable[[List[TaskInfo]], List[TaskInfo]]] = []
        self.event_handlers: List[object] = []   # objects w/ allocate/evict hooks
        # fns(tclass, job, require, forbid) mutating per-class plane bits
        self.class_constraint_hooks: List[Callable] = []
        # fns(tclass, job) -> Optional[np.ndarray [N] f32]: additive
        # per-class node score bias (task-topology bucket packing)
        self.class_bias_fns: List[Callable] = []

        # tensor-plane configuration contributed by plugins
        self.score_weights = {"least": 1.0, "most": 0.0, "bal": 0.0}
        self.dim_weights: Dict[str, float] = {}
        self.queue_limit: Optional[torch.Tensor] = None   # [Q, R]
        self.queue_alloc: Optional[torch.Tensor] = None   # [Q, R]
        self.queue_deserved: Optional[torch.Tensor] = None  # [Q, R]

        self._tier_open = False
        self.plugins: List[object] = []

    # -- tier plumbing (plugins of one tier register into the same slot) -----
    def open_tier(self) -> None:
        self.job_order_fns.append([])
        self.queue_order_fns.append([])
        self.task_order_fns.append([])
        self.job_order_keys.append([])
        self.job_order_cols.append([])
        self.queue_order_keys.append([])

    def add_job_order_fn(self, fn: CmpFn, key=None, col=None) -> None:
        """Register a job compare fn; `key` is an optional *sort key*
        equivalent (ascending) — when every registered order fn provides
        one, ordering runs as a single tuple-key sort instead of
        O(n log n) cmp callbacks (hot at 10k+ jobs).  `col` is the
        vectorized form: fn(job_table, rows) -> np.ndarray of ascending
        keys for those table rows (the columnar worksheet path)."""
        self.job_order_fns[-1].append(fn)
        self.job_order_keys[-1].append(key)
        self.job_order_cols[-1].append(col)

    def add_queue_order_fn(self, fn: CmpFn, key=None) -> None:
        self.queue_order_fns[-1].append(fn)
        self.queue_order_keys[-1].append(key)

    def add_task_order_fn(self, fn: CmpFn) -> None:
        self.task_order_fns[-1].append(fn)

    # -- aggregations (reference session_plugins.go) --------------------------
    @staticmethod
    def _tiered_cmp(tiers: List[List[CmpFn]], a, b) -> int:
        for tier in tiers:
            for fn in tier:
                r = fn(a, b)
                if r != 0:
                    return r
        return 0

    def job_order(self, a: JobInfo, b: JobInfo) -> int:
        r = self._tiered_cmp(self.job_order_fns, a, b)
        if r != 0:
            return r
        # FIFO fallback (reference: creation time then UID)
        if a.creation_timestamp != b.creation_timestamp:
            return -1 if a.creation_timestamp < b.creation_timestamp else 1
        return -1 if a.key < b.key else (1 if a.key > b.key else 0)

    def queue_order(self, a: QueueInfo, b: QueueInfo) -> int:
        r = self._tiered_cmp(self.queue_order_fns, a, b)
        if r != 0:
            return r
        return -1 if a.name < b.name else (1 if a.name > b.name else 0)

    def task_order(self, a: TaskInfo, b: TaskInfo) -> int:
        return self._tiered_cmp(self.task_order_fns, a, b)

    def job_valid(self, job: JobInfo) -> bool:
        return all(fn(job) for fn in self.job_valid_fns)

    def job_ready(self, job: JobInfo) -> bool:
        return all(fn(job) for fn in self.job_ready_fns)

    def job_pipelined(self, job: JobInfo) -> bool:
        """First non-abstain vote wins, registration order = tier order
        (reference session_plugins.go:505)."""
        for fn in self.job_pipelined_fns:
            v = fn(job)
            if v != ABSTAIN:
                return v == PERMIT
        return True

    def job_enqueueable(self, job: JobInfo) -> bool:
        """First non-abstain vote wins (reference session_plugins.go:582)."""
        for fn in self.job_enqueueable_fns:
            v = fn(job)
            if v != ABSTAIN:
                return v == PERMIT
        return True   # default permit (reference enqueue falls through)

    def job_starving(self, job: JobInfo) -> bool:
        if not self.job_starving_fns:
            return job.is_starving()
        return any(fn(job) for fn in self.job_starving_fns)

    def queue_overused(self, q: QueueInfo) -> bool:
        return any(fn(q) for fn in self.overused_fns)

    def allocatable(self, q: QueueInfo, job: JobInfo) -> bool:
        return all(fn(q, job) for fn in self.allocatable_fns)

    def preemptable(self, preemptor: TaskInfo,
                    candidates: List[TaskInfo]) -> List[TaskInfo]:
        """Intersection of victim sets across registered fns (:274)."""
        victims = candidates
        for fn in self.preemptable_fns:
            allowed = {t.uid for t in fn(preemptor, victims)}
            victims = [t for t in victims if t.uid in allowed]
            if not victims:
                return []
        return victims if self.preemptable_fns else []

    def reclaimable(self, reclaimer: TaskInfo,
                    candidates: List[TaskInfo]) -> List[TaskInfo]:
        victims = candidates
        for fn in self.reclaimable_fns:
            allowed = {t.uid for t in fn(reclaimer, victims)}
            victims = [t for t in victims if t.uid in allowed]
            if not victims:
                return []
        return victims if self.reclaimable_fns else []

    def victim_tasks(self, tasks: List[TaskInfo]) -> List[TaskInfo]:
        out: List[TaskInfo] = []
        seen = set()
        for fn in self.victim_tasks_fns:
            for t in fn(tasks):
                if t.uid not in seen:
                    seen.add(t.uid)
                    out.append(t)
        for filt in self.victim_filter_fns:
            out = filt(out)
        return out

    # -- sorted views ---------------------------------------------------------
    def sorted_queues(self, queues: Optional[List[QueueInfo]] = None) -> List[QueueInfo]:
        import functools
        qs = queues if queues is not None else list(self.queues.values())
        keys = [k for tier in self.queue_order_keys for k in tier]
        fns = [f for tier in self.queue_order_fns for f in tier]
        if len(keys) == len(fns) and all(k is not None for k in keys):
            return sorted(qs, key=lambda q: tuple(k(q) for k in keys) + (q.name,))
        return sorted(qs, key=functools.cmp_to_key(self.queue_order))

    def sorted_jobs(self, jobs: List[JobInfo]) -> List[JobInfo]:
        import functools
        keys = [k for tier in self.job_order_keys for k in tier]
        fns = [f for tier in self.job_order_fns for f in tier]
        if len(keys) == len(fns) and all(k is not None for k in keys):
            if len(jobs) >= 256:
                # columnar sort: same ordering as the tuple sort (keys
                # ascending, creation/key tie-break) without building a
                # tuple per job — np.lexsort's LAST column is primary
                import numpy as np
                cols = [np.array([k(j) for j in jobs]) for k in keys]
                ts = np.array([j.creation_timestamp for j in jobs])
                jk = np.array([j.key for j in jobs])
                order = np.lexsort(tuple([jk, ts] + cols[::-1]))
                return [jobs[i] for i in order]
            return sorted(jobs, key=lambda j: tuple(k(j) for k in keys)
                          + (j.creation_timestamp, j.key))
        return sorted(jobs, key=functools.cmp_to_key(self.job_order))

    def ordered_job_rows(self, jt, rows):
        """Vectorized job ordering over JobTable rows: same ordering as
        ``sorted_jobs`` (tier keys ascending, creation/key tie-break) but
        computed by np.lexsort over column providers."""
        import numpy as np
        if len(rows) <= 1:
            return rows
        keys = [k for tier in self.job_order_keys for k in tier]
        fns = [f for tier in self.job_order_fns for f in tier]
        cols = [c for tier in self.job_order_cols for c in tier]
        if len(keys) != len(fns) or any(k is None for k in keys):
            jobs = [jt.jobs[int(i)] for i in rows]
            pos = {id(j): r for j, r in zip(jobs, rows)}
            return np.array([pos[id(j)] for j in self.sorted_jobs(jobs)],
                            dtype=rows.dtype)
        arrs = []
        for k, c in zip(keys, cols):
            if c is not None:
                arrs.append(np.asarray(c(jt, rows)))
            else:
                arrs.append(np.array([k(jt.jobs[int(i)]) for i in rows]))
        # sigid sits BELOW every plugin key and ABOVE ctime: jobs the
        # plugins left tied group by plan-atom signature, so identical
        # gangs run consecutively and the allocate bundler fuses them
        # into single kernel passes (heterogeneous-mix shape).  Legal:
        # the reference's sort.Slice leaves tie order unspecified.
        order = np.lexsort(tuple([jt.keys[rows], jt.ctime[rows],
                                  jt.sigid[rows]] + arrs[::-1]))
        return rows[order]

    # -- queue tensor rows ----------------------------------------------------
    def build_queue_tensors(self) -> None:
        """queue_alloc from currently-allocated tasks; limit defaults open.
        Plugins (proportion/capacity) overwrite limit rows at session open."""
        import numpy as np
        nt = self.node_tensors
        Q, R = len(self.queues), nt.r
        if not self.queue_index or len(self.queue_index) != Q:
            self.queue_index = {name: i
                                for i, name in enumerate(sorted(self.queues))}
        alloc = np.zeros((Q, R), dtype=np.float32)
        jt = getattr(self, "job_table", None)
        if jt is not None and len(jt.jobs) == len(self.jobs):
            rows = np.nonzero(jt.occ > 0)[0]
            for k in rows:
                job = jt.jobs[k]
                qi = jt.qi[k]
                if qi >= 0:
                    alloc[qi] += job.alloc_vec(nt)
        else:
            for job in self.jobs.values():
                if job.occupied_count == 0:
                    continue      # nothing allocated — skip the vector build
                qi = self.queue_index.get(job.queue)
                if qi is not None:
                    alloc[qi] += job.alloc_vec(nt)
        self.queue_alloc = torch.from_numpy(alloc)
        self.queue_limit = torch.full((Q, R), BIG_LIMIT, dtype=torch.float32)

    def dim_weight_vector(self) -> torch.Tensor:
        nt = self.node_tensors
        w = torch.ones(nt.r, dtype=torch.float32)
        for name, val in self.dim_weights.items():
            i = nt.dims.index.get(name)
            if i is not None:
                w[i] = val
        return w

    # -- events ---------------------------------------------------------------
    def fire_allocate(self, task_class, node_ids, counts,
                      tasks: Optional[List[TaskInfo]] = None) -> None:
        for h in self.event_handlers:
            fn = getattr(h, "on_allocate", None)
            if fn:
                fn(task_class, node_ids, counts, tasks)

    def fire_evict(self, task: TaskInfo) -> None:
        for h in self.event_handlers:
            fn = getattr(h, "on_evict", None)
            if fn:
                fn(task)
