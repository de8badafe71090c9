"""Scheduler cache: incremental mirror of the store + bind pipeline.

Analog of ``pkg/scheduler/cache/`` (SchedulerCache, event_handlers.go,
interface.go Binder/Evictor).  Key MI355X-first departure: the reference
deep-clones the whole world every cycle (cache.go:1481 Snapshot — its
dominant fixed cost, SURVEY.md §7 hard-parts); here the cache applies
store watch deltas incrementally to persistent infos, and the per-cycle
"snapshot" is just a tensor re-pack of node planes (the session borrows
the live infos — the cycle is single-threaded by design, so there is
nothing to clone; the reference needs the clone because its informers
mutate concurrently).
"""

from __future__ import annotations

import threading
from typing import Dict, List, Optional

from ..api.info import JobInfo, NodeInfo, QueueInfo, TaskInfo
from ..api.objects import ObjectMeta, PodGroup, Queue
from ..api.resource import ResourceDims
from ..api.types import TaskStatus
from ..store import EventType, ObjectStore
from ..utils.features import enabled as features_enabled
from .tensors import NodeTensors


class Binder:
    """Commit interface (reference cache/interface.go:113-120)."""

    def bind(self, tasks: List[TaskInfo]) -> None:  # pragma: no cover
        raise NotImplementedError

    def evict(self, task: TaskInfo, reason: str = "") -> None:  # pragma: no cover
        raise NotImplementedError


class FakeBinder(Binder):
    """Records binds/evictions (reference util/test_utils.go:536 FakeBinder)
    — also the bench-mode sink so the GPU loop, not a cluster API, is
    measured (SURVEY.md §7 'binding throughput').

    Intake is batched: ``bind`` stores the task list; the ``binds`` dict
    materializes lazily on read (tests) so the hot path pays no per-task
    dict stores.  ``bound_count`` is the O(1) bench-side counter."""

    def __init__(self):
        self._binds: Dict[str, str] = {}
        self._spans: List[List[TaskInfo]] = []
        self.bound_count = 0
        self.evictions: List[str] = []

    @property
    def binds(self) -> Dict[str, str]:
        if self._spans:
            d = self._binds
            for ts in self._spans:
                for t in ts:
                    d[t.key] = t.node_name
            self._spans.clear()
        return self._binds

    def bind(self, tasks: List[TaskInfo]) -> None:
        self._spans.append(tasks)
        self.bound_count += len(tasks)

    def clear(self) -> None:
        self._binds.clear()
        self._spans.clear()
        self.bound_count = 0

    def evict(self, task: TaskInfo, reason: str = "") -> None:
        self.evictions.append(task.key)


class StoreBinder(Binder):
    """Writes the binding back to the object store (the apiserver analog)."""

    def __init__(self, store: ObjectStore):
        self.store = store

    def bind(self, tasks: List[TaskInfo]) -> None:
        for t in tasks:
            pod = self.store.get("Pod", t.namespace, t.name)
            if pod is not None:
                pod.node_name = t.node_name
                if pod.volumes and features_enabled("CSIStorage"):
                    self._bind_volumes(pod)
                self.store.update("Pod", pod)

    def _bind_volumes(self, pod) -> None:
        """Schedule-time volume binding (k8s volumebinding's assume step,
        WaitForFirstConsumer): each of the pod's still-unbound claims gets
        a free PV that fits — capacity, storage class, and the zone of the
        node the pod just landed on."""
        from ..api.objects import ZONE_LABEL
        node = self.store.get("Node", "default", pod.node_name) or \
            next(iter(self.store.list(
                "Node", selector=lambda n: n.meta.name == pod.node_name)),
                None)
        node_zone = node.meta.labels.get(ZONE_LABEL, "") if node else ""
        free = None
        for vname in pod.volumes:
            pvc = self.store.get("PersistentVolumeClaim",
                                 pod.meta.namespace, vname)
            if pvc is None or pvc.volume_name:
                continue
            if free is None:
                free = sorted((pv for pv in self.store.list("PersistentVolume")
                               if not pv.claim_ref),
                              key=lambda pv: (pv.capacity, pv.meta.name))
            for i, pv in enumerate(free):
                if pv.capacity < pvc.request:
                    continue
                if pvc.storage_class and pv.storage_class != pvc.storage_class:
                    continue
                if pv.zone and node_zone and pv.zone != node_zone:
                    continue
                pv.claim_ref = pvc.meta.key
                pvc.volume_name = pv.meta.name
                self.store.update("PersistentVolume", pv)
                self.store.update("PersistentVolumeClaim", pvc)
                free.pop(i)
                break

    def evict(self, task: TaskInfo, reason: str = "") -> None:
        pod = self.store.get("Pod", task.namespace, task.name)
        if pod is not None:
            pod.phase = "Failed"
            pod.meta.annotations["volcano.sh/evicted"] = reason or "preempted"
            self.store.update("Pod", pod)


class SchedulerCache:
    def __init__(self, store: Optional[ObjectStore] = None,
                 binder: Optional[Binder] = None, device: str = "cpu",
                 scheduler_name: str = "volcano"):
        self.store = store
        self.binder = binder or (StoreBinder(store) if store else FakeBinder())
        self.scheduler_name = scheduler_name
        self.dims = ResourceDims()
        self.device = device
        self.node_tensors = NodeTensors(self.dims, device=device)

        self.jobs: Dict[str, JobInfo] = {}
        self.jobs_epoch = 0                     # bumped on job add/remove
        from .jobtable import JobTable
        self.job_table = JobTable()
        self.nodes: Dict[str, NodeInfo] = {}
        self.queues: Dict[str, QueueInfo] = {}
        self._task_node: Dict[str, str] = {}    # task key -> node name
        self._task_job: Dict[str, str] = {}     # task key -> job key
        self._task_ref: Dict[str, TaskInfo] = {}  # task key -> live TaskInfo
        self._lock = threading.RLock()
        self._watch = store.watch("Pod", "Node", "PodGroup", "Queue") \
            if store else None
        self._tensors_dirty = True      # full repack (node set/labels changed)
        self._used_dirty = False        # dynamic planes only (usage changed)
        # jobs holding PIPELINED reservations (noted by Statement.pipeline)
        # — _demote_pipelined scans only these instead of every job
        self._pipelined_jobs: set = set()
        self.had_nominations = False

    # -- event ingestion (reference cache/event_handlers.go) -----------------
    def sync(self) -> int:
        """Drain pending store events into the infos; returns event count."""
        if self._watch is None:
            return 0
        if self._watch.overflowed:
            return self._resync()
        evs = self._watch.drain()
        # within a batch, structure before pods (a replayed Pod may refer to
        # a Node/PodGroup whose ADDED event is later in the same drain)
        kind_rank = {"Node": 0, "Queue": 1, "PodGroup": 2, "Pod": 3}
        evs.sort(key=lambda e: kind_rank.get(e.kind, 4))
        with self._lock:
            for ev in evs:
                handler = getattr(self, f"_on_{ev.kind.lower()}", None)
                if handler:
                    handler(ev)
            if evs:
                if any(ev.kind == "Node" for ev in evs):
                    self._tensors_dirty = True      # static planes changed
                else:
                    self._used_dirty = True         # usage-only delta
        return len(evs)

    def _resync(self) -> int:
        """Watch overflowed (events dropped under backpressure): discard
        the stale stream and rebuild every info from the store's current
        state — the informer relist path (k8s 410 Gone → re-list)."""
        with self._lock:
            self._watch.stop()
            self.jobs.clear()
            self.nodes.clear()
            self.queues.clear()
            self._task_node.clear()
            self._task_job.clear()
            self._task_ref.clear()
            self._watch = self.store.watch("Pod", "Node", "PodGroup", "Queue")
            self._tensors_dirty = True
            self.jobs_epoch += 1
        return self.sync()

    def _job_for(self, pg_key: str) -> JobInfo:
        job = self.jobs.get(pg_key)
        if job is None:
            job = self.jobs[pg_key] = JobInfo(pg_key)
            self.jobs_epoch += 1
        return job

    def _on_pod(self, ev) -> None:
        pod = ev.obj
        if pod.scheduler_name != self.scheduler_name:
            return
        pg = pod.podgroup_name
        if not pg:
            # normal pod without a group: implicit single-pod group
            pg = f"pod-{pod.meta.name}"
        key = f"{pod.meta.namespace}/{pg}"
        tkey = pod.meta.key

        # remove any previous incarnation — node accounting keys off the
        # tracked TaskInfo object, NOT the job lookup: a PodGroup DELETE in
        # the same batch may have dropped the job already, and the node's
        # usage must still unwind (found by tests/test_churn_e2e.py)
        old_node = self._task_node.pop(tkey, None)
        old_job = self._task_job.pop(tkey, None)
        old_task = self._task_ref.pop(tkey, None)
        if old_job is not None and old_job in self.jobs:
            self.jobs[old_job].remove_task(tkey)
        if old_task is not None:
            if old_node and old_node in self.nodes:
                self.nodes[old_node].remove_task(old_task)
            if ev.type == EventType.DELETED:
                self._release_devices(old_task, old_node)

        if ev.type == EventType.DELETED:
            return
        # DRA: resolve the pod's ResourceClaims into dra:<class> request
        # dims (predicates + queue quotas then ride the dense machinery)
        if pod.resource_claims and self.store is not None:
            for cname in pod.resource_claims:
                claim = self.store.get("ResourceClaim",
                                       pod.meta.namespace, cname)
                if claim is not None and claim.device_class_name:
                    dim = f"dra:{claim.device_class_name}"
                    pod.request.q[dim] = pod.request.q.get(dim, 0.0) + \
                        float(claim.count)
        task = TaskInfo.from_pod(pod, key)
        for k in task.request.q:
            if k.startswith(("paa:", "hp:", "nvl:", "dra:")) \
                    and k not in self.dims:
                self.dims.add(k)
                self._tensors_dirty = True
        self._job_for(key).add_task(task)
        self._task_job[tkey] = key
        self._task_ref[tkey] = task
        if task.node_name and task.node_name in self.nodes:
            self.nodes[task.node_name].add_task(task)
            self._task_node[tkey] = task.node_name

    def _release_devices(self, task: TaskInfo, node_name: Optional[str]) -> None:
        """Return GPU slices to the node pool when a device pod goes away
        (deviceshare plugin owns allocation; see api/devices.py)."""
        pools = getattr(self, "_device_pools", None)
        if not pools or task.pod is None or not node_name:
            return
        from ..api.devices import ANN_ASSIGNED, DeviceRequest
        assigned = task.pod.meta.annotations.get(ANN_ASSIGNED)
        req = DeviceRequest.from_annotations(task.pod.meta.annotations)
        pool = pools.get(node_name)
        if assigned and req is not None and pool is not None:
            pool.release(req, [int(x) for x in assigned.split(",") if x])

    def _on_node(self, ev) -> None:
        name = ev.obj.meta.name
        if ev.type == EventType.DELETED:
            self.nodes.pop(name, None)
            return
        old = self.nodes.get(name)
        ni = NodeInfo(ev.obj)
        if old is not None:
            for t in old.tasks.values():
                ni.add_task(t)
            ni.remote_used = old.remote_used.clone()
            ni.used.add(ni.remote_used)
        self.nodes[name] = ni

    def _on_podgroup(self, ev) -> None:
        pg: PodGroup = ev.obj
        key = pg.meta.key
        if ev.type == EventType.DELETED:
            if self.jobs.pop(key, None) is not None:
                self.jobs_epoch += 1
            return
        job = self._job_for(key)
        job.podgroup = pg
        job._minav = None      # minMember may have changed
        job._mtm = None
        job._tver += 1

    def _on_queue(self, ev) -> None:
        q: Queue = ev.obj
        if ev.type == EventType.DELETED:
            self.queues.pop(q.meta.name, None)
            return
        self.queues[q.meta.name] = QueueInfo(q)

    # -- direct population (uthelper-style tests / bench) ---------------------
    def add_node_info(self, ni: NodeInfo) -> None:
        self.nodes[ni.name] = ni
        self._tensors_dirty = True

    def add_job_info(self, job: JobInfo) -> None:
        self.jobs[job.key] = job
        self.jobs_epoch += 1
        for t in job.tasks.values():
            if t.node_name and t.node_name in self.nodes:
                self.nodes[t.node_name].add_task(t)
        self._used_dirty = True

    def add_queue_info(self, qi: QueueInfo) -> None:
        self.queues[qi.name] = qi

    # -- snapshot -------------------------------------------------------------
    def snapshot_into(self, ssn) -> None:
        self.sync()
        self.had_nominations = False
        self._demote_pipelined()
        if "default" not in self.queues:
            self.queues["default"] = QueueInfo(
                Queue(meta=ObjectMeta(name="default")))
        self.ensure_packed()
        # dynamic node-set bits are per-session; plugins re-project them
        self.node_tensors.clear_dynamic_bits()
        ssn.jobs = self.jobs
        ssn.nodes = self.nodes
        ssn.queues = self.queues
        ssn.node_tensors = self.node_tensors
        ssn.total_resource = self.node_tensors.alloc_t.sum(dim=1).to("cpu")
        ssn.queue_index = {name: i
                           for i, name in enumerate(sorted(self.queues))}
        self.job_table.refresh(self, self.node_tensors, ssn.queue_index)
        ssn.job_table = self.job_table

    def note_pipelined(self, job_key: str) -> None:
        self._pipelined_jobs.add(job_key)

    def _demote_pipelined(self) -> None:
        """Pipelined reservations live one cycle: at the next snapshot they
        re-enter Pending and compete with the capacity their evictions
        freed.  (The reference carries Pipelined across cycles through
        PodGroupOldState — session.go:77-79; the one-cycle reservation is
        the plan-design equivalent: the gang stays pipelined through the
        session in which its evictions were committed, and converges to
        Bound the following cycle.)  Only jobs noted by
        Statement.pipeline are scanned (plus none at all in the common
        no-preemption cycle)."""
        if not self._pipelined_jobs:
            return
        noted, self._pipelined_jobs = self._pipelined_jobs, set()
        for key in noted:
            job = self.jobs.get(key)
            if job is None:
                continue
            bucket = job.task_status_index.get(TaskStatus.PIPELINED)
            if not bucket:
                continue
            pipelined = list(bucket.values())
            for t in pipelined:
                node = self.nodes.get(t.node_name)
                if node is not None:
                    node.remove_task(t)
                t.nominated_node = t.node_name   # allocate fast-path hint
                t.node_name = ""
                job.update_task_status(t, TaskStatus.PENDING)
            if pipelined:
                job._nom = True       # allocate checks this before the
                self.had_nominations = True
                self._used_dirty = True

    def ensure_packed(self) -> None:
        nodes = sorted(self.nodes.values(), key=lambda n: n.name)
        # node_id-ordered list for apply/preempt paths (avoids re-sorting
        # 10k+ NodeInfos once per action)
        self.nodes_sorted = nodes
        if self._tensors_dirty or self.node_tensors.alloc_t is None:
            # (re)adopt the node set into the columnar usage ledger first
            # so the pack reads usage as one vectorized plane copy
            from ..api.ledger import NodeLedger
            self.ledger = NodeLedger.build(self.dims, nodes)
            self.node_tensors.pack(nodes, self.ledger)
            self._tensors_dirty = self._used_dirty = False
            for i, ni in enumerate(nodes):
                ni.node_id = i
        elif self._used_dirty:
            if not self.node_tensors.pack_dynamic(
                    nodes, getattr(self, "ledger", None)):
                from ..api.ledger import NodeLedger
                self.ledger = NodeLedger.build(self.dims, nodes)
                self.node_tensors.pack(nodes, self.ledger)
                for i, ni in enumerate(nodes):
                    ni.node_id = i
            self._tensors_dirty = self._used_dirty = False

    def reset_usage(self) -> None:
        """Zero every node's usage accounting and task membership in one
        vectorized pass (bench/soak step reset; also the crash-resume
        rebuild path).  The ledger planes zero wholesale; unadopted nodes
        fall back to fresh local Resources."""
        from ..api.resource import Resource
        led = getattr(self, "ledger", None)
        if led is not None:
            led.zero_usage()
        for ni in self.nodes.values():
            if ni._ledger is None or ni._row < 0:
                ni._used = Resource()
                ni._releasing = Resource()
                ni._pipelined = Resource()
                ni._remote_used = Resource()
            ni._tasks.clear()
            ni._batches.clear()
        self._used_dirty = True

    # -- commit pipeline ------------------------------------------------------
    def bind_tasks(self, tasks: Optional[List[TaskInfo]],
                   by_job: Optional[Dict[str, List[TaskInfo]]] = None,
                   preset: bool = False) -> None:
        """Async in the reference (cache.go:1343 AddBindTask → 20 ms drain);
        here a batched call — the binder itself may thread if it wants.
        Callers that already have the per-job grouping may pass ONLY
        by_job (tasks=None): the flat list is not materialized."""
        if by_job is None:
            by_job = {}
            for t in tasks:
                by_job.setdefault(t.job_key, []).append(t)
        if tasks is None:
            if self._watch is not None:
                for ts in by_job.values():
                    self._task_node.update((t.key, t.node_name)
                                           for t in ts)
            for ts in by_job.values():
                self.binder.bind(ts)
        else:
            if self._watch is not None:
                # incarnation bookkeeping only matters when store events flow
                self._task_node.update((t.key, t.node_name) for t in tasks)
            self.binder.bind(tasks)
        # ``preset``: the caller's task walk already wrote BOUND statuses
        # (allocate._apply fuses it with the node_name pass); only the
        # bucket move remains
        for key, ts in by_job.items():
            job = self.jobs.get(key)
            if job is not None:
                if preset:
                    job.finish_bind(ts)
                else:
                    job.move_tasks_status(ts, TaskStatus.BOUND)

    def evict_task(self, task: TaskInfo, reason: str = "") -> None:
        self.binder.evict(task, reason)
        job = self.jobs.get(task.job_key)
        node = self.nodes.get(task.node_name)
        # remove BEFORE the status flip (remove subtracts by the task's
        # CURRENT status buckets), then re-add as RELEASING so the node's
        # releasing total is credited — same order as Statement.evict
        if node is not None:
            node.remove_task(task)
        if job is not None:
            job.update_task_status(task, TaskStatus.RELEASING)
        else:
            task.status = TaskStatus.RELEASING
        if node is not None:
            node.add_task(task)
        self._used_dirty = True

    def update_podgroup(self, job: JobInfo) -> None:
        if self.store is not None and job.podgroup is not None:
            try:
                self.store.update("PodGroup", job.podgroup)
            except KeyError:
                pass
