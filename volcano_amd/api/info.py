"""Scheduler in-memory domain model: TaskInfo / JobInfo / NodeInfo / QueueInfo.

Analog of the reference's ``pkg/scheduler/api/{job_info,node_info,
queue_info}.go`` — but reshaped for tensor batching: tasks with identical
requests *and* identical scheduling constraints are grouped into a
``TaskClass`` so one HIP kernel pass places a whole class (the reference
predicates/scores every (task, node) pair one task at a time,
util/predicate_helper.go:45).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from .objects import (ANN_PREEMPTABLE, LBL_TASK_SPEC, Pod, Node, PodGroup,
                      Queue)
from .resource import Resource
from .types import ALLOCATED_STATUSES, PodGroupPhase, TaskStatus


@dataclass(slots=True)
class TaskInfo:
    """One schedulable pod (reference api/job_info.go:118-175).
    Slots: at 100k+ live tasks the per-instance dict is the dominant
    host-memory cost and attribute access is the hottest interpreter op
    in the commit path."""

    uid: str
    name: str
    namespace: str
    job_key: str
    role: str                      # task-spec name (volcano.sh/task-spec)
    request: Resource
    status: TaskStatus = TaskStatus.PENDING
    node_name: str = ""
    priority: int = 0
    best_effort: bool = False
    preemptable: bool = False
    revocable_zone: str = ""
    gated: bool = False            # has scheduling gates (k8s SchedulingGates)
    # node a preemption pipelined this task onto last cycle — the allocate
    # fast path re-checks fit there before full scoring (reference
    # NominatedNodeName, allocate.go:797)
    nominated_node: str = ""
    pod: Optional[Pod] = None
    # lazy caches (slots-compatible; excluded from init/repr/compare)
    _key: Optional[str] = field(default=None, init=False, repr=False,
                                compare=False)
    _sig: Optional[tuple] = field(default=None, init=False, repr=False,
                                  compare=False)

    @property
    def key(self) -> str:
        k = self._key
        if k is None:
            k = self._key = f"{self.namespace}/{self.name}"
        return k

    def class_signature(self):
        """Tasks with equal signatures are scheduled as one batch.
        Memoized — constraints are immutable for a task's lifetime.
        A plain hashable tuple (not a digest: hashing+json per task was
        the dominant cold-snapshot cost at 100k pods).  Distinct
        constraints always yield distinct tuples; `aff` uses repr so two
        equal-but-differently-ordered dicts may split a batch (safe:
        batching is an optimization, never a correctness input)."""
        sig = self._sig
        if sig is not None:
            return sig
        p = self.pod
        # empty-guard every optional facet: the common pod (no selector,
        # tolerations, gates, device annotations or volumes) skips all
        # eight sorted()/genexpr constructions — the signature build was
        # the dominant arrival-path cost under sustained churn
        if p is not None:
            sel = p.node_selector
            tol = p.tolerations
            gates = p.scheduling_gates
            anns = p.meta.annotations
            vols = p.volumes
            sig = (
                self.role,
                self.preemptable,   # tdm/rescheduling constraints differ
                tuple(sorted(self.request.q.items())),
                tuple(sorted(sel.items())) if sel else (),
                tuple((t.key, t.operator, t.value, t.effect)
                      for t in tol) if tol else (),
                repr(p.affinity) if p.affinity else "",
                self.priority,
                tuple(sorted(gates)) if gates else (),
                tuple(sorted(k for k in anns
                             if k.startswith("volcano.sh/gpu")
                             or k.startswith("volcano.sh/vgpu")
                             or k == "volcano.sh/numa-topology-policy"))
                if anns else (),
                tuple(sorted(vols)) if vols else (),
            )
        else:
            sig = (
                self.role,
                self.preemptable,
                tuple(sorted(self.request.q.items())),
                (), (), "", self.priority, (), (), (),
            )
        # intern: equal signatures share ONE tuple object, so downstream
        # equality checks (bundle-continuation keys, memo dicts) hit the
        # identity fast path instead of deep tuple compares — measurable
        # at 10k identical gangs per cycle.  Bounded by distinct classes.
        sig = _SIG_INTERN.setdefault(sig, sig)
        self._sig = sig
        return sig

    @classmethod
    def from_pod(cls, pod: Pod, job_key: str) -> "TaskInfo":
        status = {
            "Pending": TaskStatus.PENDING if not pod.node_name else TaskStatus.BOUND,
            "Running": TaskStatus.RUNNING,
            "Succeeded": TaskStatus.SUCCEEDED,
            "Failed": TaskStatus.FAILED,
        }.get(pod.phase, TaskStatus.UNKNOWN)
        # pod anti-affinity groups ride the resource machinery: group G is
        # a synthetic unit dim (every node offers 1 — tensors.py pack), so
        # the capacity kernel enforces at-most-one-per-node with zero
        # extra predicate work (plugins/interpodaffinity.py)
        aff = pod.affinity if isinstance(pod.affinity, dict) else None
        if aff:
            anti = aff.get("podAntiAffinity")
            if isinstance(anti, dict) and anti.get("group"):
                pod.request.q.setdefault(f"paa:{anti['group']}", 1.0)
        # host ports and attachable volumes ride the same machinery: each
        # port is a per-node unit dim (k8s nodeports filter), the volume
        # count draws from the node's attach budget (nodevolumelimits)
        for p in pod.host_ports:
            pod.request.q.setdefault(f"hp:{int(p)}", 1.0)
        if pod.volumes:
            pod.request.q.setdefault("nvl:attach", float(len(pod.volumes)))
        return cls(
            uid=pod.meta.uid or pod.meta.key,
            name=pod.meta.name,
            namespace=pod.meta.namespace,
            job_key=job_key,
            role=pod.meta.labels.get(LBL_TASK_SPEC, ""),
            request=pod.request.clone(),
            status=status,
            node_name=pod.node_name,
            priority=pod.priority,
            best_effort=pod.best_effort,
            # reference GetPodPreemptable (api/pod_info.go:176): explicit
            # annotation wins, label fallback, DEFAULT TRUE
            preemptable=_pod_preemptable(pod),
            gated=bool(pod.scheduling_gates),
            pod=pod,
        )


@dataclass(slots=True)
class TaskClass:
    """A batch of identical pending tasks of one job — the kernel launch unit."""

    signature: str
    role: str
    request: Resource
    tasks: List[TaskInfo] = field(default_factory=list)
    priority: int = 0
    # filled by the snapshot packer:
    class_id: int = -1
    # per-class network-topology override (SubGroupPolicy): the
    # network-topology-aware plugin confines THIS class to one domain
    # instead of using the job-level spec
    topology: Optional[dict] = None

    @property
    def count(self) -> int:
        return len(self.tasks)


def _pod_preemptable(pod) -> bool:
    val = pod.meta.annotations.get(ANN_PREEMPTABLE)
    if val is None:
        val = pod.meta.labels.get(ANN_PREEMPTABLE)
    return True if val is None else val == "true"


_OCCUPIED_STATUSES = tuple(ALLOCATED_STATUSES) + (TaskStatus.SUCCEEDED,)
_OCC_SET = frozenset(_OCCUPIED_STATUSES)
_SIG_INTERN: Dict[tuple, tuple] = {}


class JobInfo:
    """A PodGroup plus its tasks (reference api/job_info.go:399-445)."""

    def __init__(self, key: str, podgroup: Optional[PodGroup] = None):
        self.key = key                       # namespace/name of the podgroup
        self.podgroup = podgroup
        self.tasks: Dict[str, TaskInfo] = {}           # task key → info
        self.task_status_index: Dict[TaskStatus, Dict[str, TaskInfo]] = {}
        self.job_id = -1                     # dense index assigned by snapshot
        # dense-vector caches (invalidated on task/status mutation)
        self._alloc_vec = None               # (r, np.ndarray)
        self._total_vec = None
        # plan atom: (sig, role, request, priority) when the job's task
        # set is a single class; False = known multi-class; None = unknown
        self._atom = None
        self._minav = None
        self._occ = 0                        # occupied-count (incremental)
        self._tver = 0                       # static-shape version (JobTable)
        self._jrow = -1                      # row in the cycle JobTable
        self._mtm = None                     # cached minTaskMember map
        self._nom = False                    # any task carries a nomination

    # -- basic accessors ----------------------------------------------------
    @property
    def name(self) -> str:
        return self.key.split("/", 1)[1]

    @property
    def namespace(self) -> str:
        return self.key.split("/", 1)[0]

    @property
    def queue(self) -> str:
        return self.podgroup.spec.queue if self.podgroup else "default"

    @property
    def min_available(self) -> int:
        m = self._minav
        if m is None:
            m = self._minav = \
                self.podgroup.spec.min_member if self.podgroup else 1
        return m

    @property
    def min_task_member(self) -> Dict[str, int]:
        m = self._mtm
        if m is None:
            m = self._mtm = \
                self.podgroup.spec.min_task_member if self.podgroup else {}
        return m

    @property
    def priority(self) -> int:
        p = getattr(self, "_prio", None)
        if p is not None:
            return p
        if self.podgroup and self.podgroup.meta.annotations.get("priority"):
            p = int(self.podgroup.meta.annotations["priority"])
        else:
            p = max((t.priority for t in self.tasks.values()), default=0)
        self._prio = p
        return p

    @property
    def creation_timestamp(self) -> float:
        return self.podgroup.meta.creation_timestamp if self.podgroup else 0.0

    @property
    def phase(self) -> str:
        return self.podgroup.status.phase if self.podgroup else PodGroupPhase.PENDING.value

    # -- task bookkeeping ---------------------------------------------------
    def add_task(self, task: TaskInfo) -> None:
        self.tasks[task.key] = task
        self.task_status_index.setdefault(task.status, {})[task.key] = task
        if task.status in _OCC_SET:
            self._occ += 1
        self._alloc_vec = self._total_vec = self._atom = None
        self._prio = None
        self._tver += 1
        if task.gated:
            self._gated = getattr(self, "_gated", 0) + 1

    @property
    def has_gated_tasks(self) -> bool:
        return getattr(self, "_gated", 0) > 0

    def remove_task(self, task_key: str) -> Optional[TaskInfo]:
        t = self.tasks.pop(task_key, None)
        if t is not None:
            self.task_status_index.get(t.status, {}).pop(task_key, None)
            if t.status in _OCC_SET:
                self._occ -= 1
            self._alloc_vec = self._total_vec = self._atom = None
            self._tver += 1
        return t

    def plan_atom(self):
        """(sig, role, request, priority) if every task of the job forms
        ONE scheduling class — the steady-state fast path of the allocate
        plan builder (uniform gangs).  False when known multi-class."""
        if self._atom is not None:
            return self._atom
        it = iter(self.tasks.values())
        first = next(it, None)
        if first is None:
            self._atom = False
            return False
        sig = first.class_signature()
        for t in it:
            if t.class_signature() != sig:
                self._atom = False
                return False
        self._atom = (sig, first.role, first.request, first.priority)
        return self._atom

    def update_task_status(self, task: TaskInfo, status: TaskStatus) -> None:
        self.task_status_index.get(task.status, {}).pop(task.key, None)
        self._occ += (status in _OCC_SET) - (task.status in _OCC_SET)
        task.status = status
        self.task_status_index.setdefault(status, {})[task.key] = task
        self._alloc_vec = None

    def move_tasks_status(self, tasks: List[TaskInfo],
                          status: TaskStatus) -> None:
        """Bulk status move (hot path: a whole class commits at once).
        When the batch is an entire status bucket (the common gang case:
        all pending → bound), the bucket dict moves wholesale."""
        if tasks:
            # callers pass tasks drawn from one status bucket (plan apply /
            # bind); when the batch covers the whole bucket it moves
            # wholesale.  Size equality implies identity here because the
            # batch is a subset of the bucket by construction.
            src_status = tasks[0].status
            src = self.task_status_index.get(src_status)
            if src is not None and len(src) == len(tasks):
                self._occ += ((status in _OCC_SET) -
                              (src_status in _OCC_SET)) * len(tasks)
                for t in tasks:
                    t.status = status
                dst = self.task_status_index.get(status)
                if dst:
                    dst.update(src)
                else:
                    self.task_status_index[status] = src
                self.task_status_index[src_status] = {}
                self._alloc_vec = None
                return
        dst = self.task_status_index.setdefault(status, {})
        in_occ = status in _OCC_SET
        for t in tasks:
            self.task_status_index.get(t.status, {}).pop(t.key, None)
            self._occ += in_occ - (t.status in _OCC_SET)
            t.status = status
            dst[t.key] = t
        self._alloc_vec = None

    def finish_bind(self, tasks: List[TaskInfo]) -> None:
        """Index repair after a caller-fused bind: the apply walk already
        set ``t.status = BOUND`` on formerly-PENDING tasks in the same
        pass that set ``node_name`` (one task walk instead of two).  Only
        the bucket move and occupancy remain — wholesale when the batch
        is the whole pending bucket (the gang case)."""
        src = self.task_status_index.get(TaskStatus.PENDING)
        n = len(tasks)
        if src is not None and len(src) == n:
            dst = self.task_status_index.get(TaskStatus.BOUND)
            if dst:
                dst.update(src)
            else:
                self.task_status_index[TaskStatus.BOUND] = src
            self.task_status_index[TaskStatus.PENDING] = {}
        else:
            dst = self.task_status_index.setdefault(TaskStatus.BOUND, {})
            for t in tasks:
                if src is not None:
                    src.pop(t.key, None)
                dst[t.key] = t
        self._occ += n          # BOUND occupies, PENDING does not
        self._alloc_vec = None

    # -- dense vector caches (hot: plugin tensor builds) ---------------------
    def alloc_vec(self, nt):
        """Allocated-resource vector over nt.dims (cached)."""
        if self._alloc_vec is not None and self._alloc_vec[0] == nt.r:
            return self._alloc_vec[1]
        vec = nt.resource_vector(self.allocated_resource())
        self._alloc_vec = (nt.r, vec)
        return vec

    def total_req_vec(self, nt):
        """Total-request vector over nt.dims (cached)."""
        if self._total_vec is not None and self._total_vec[0] == nt.r:
            return self._total_vec[1]
        vec = nt.resource_vector(self.total_request())
        self._total_vec = (nt.r, vec)
        return vec

    def minres_vec(self, nt):
        """PodGroup minResources vector (immutable per podgroup spec)."""
        pg = self.podgroup
        cached = getattr(self, "_minres_vec_c", None)
        if cached is not None and cached[0] == nt.r:
            return cached[1]
        import numpy as np
        vec = nt.resource_vector(pg.spec.min_resources) if pg is not None \
            else np.zeros(nt.r, dtype=np.float32)
        self._minres_vec_c = (nt.r, vec)
        return vec

    def minres_unrepresented(self, nt) -> bool:
        """True if minResources demands a resource NO dim represents
        (nothing in the cluster offers or accounts it).  Dense vectors
        silently drop unknown dims, so admission gates must reject these
        explicitly — the reference computes realCapability 0 for an
        unknown resource and refuses the enqueue (proportion.go
        jobEnqueueableFn minReq ≤ realCapability)."""
        pg = self.podgroup
        if pg is None or not pg.spec.min_resources.q:
            return False
        didx = nt.dims.index
        for k, v in pg.spec.min_resources.q.items():
            if v > 0 and k not in didx:
                return True
        return False

    def tasks_with_status(self, *statuses: TaskStatus) -> List[TaskInfo]:
        out: List[TaskInfo] = []
        for s in statuses:
            out.extend(self.task_status_index.get(s, {}).values())
        return out

    # -- derived quantities -------------------------------------------------
    @property
    def pending_tasks(self) -> List[TaskInfo]:
        return self.tasks_with_status(TaskStatus.PENDING)

    @property
    def occupied_count(self) -> int:
        """Tasks holding or promised resources (reference ReadyTaskNum:
        Bound+Binding+Running+Allocated+Succeeded).  Maintained
        incrementally by the four mutation methods — this is the hottest
        job property (gang checks, ordering keys, queue tensors)."""
        return self._occ

    @property
    def waiting_count(self) -> int:
        return len(self.task_status_index.get(TaskStatus.PIPELINED, ()))

    def is_ready(self) -> bool:
        return self.occupied_count >= self.min_available

    def is_pipelined(self) -> bool:
        return self.occupied_count + self.waiting_count >= self.min_available

    def is_starving(self) -> bool:
        return self.occupied_count + self.waiting_count < self.min_available

    def role_occupied(self, role: str) -> int:
        return sum(1 for t in self.tasks.values()
                   if t.role == role and (t.status.occupies_node or t.status == TaskStatus.SUCCEEDED))

    def roles_ready(self) -> bool:
        """Per-role minimums (minTaskMember) — reference CheckTaskReady."""
        mtm = self.min_task_member
        if not mtm:
            return True
        for role, need in mtm.items():
            if self.role_occupied(role) < need:
                return False
        return True

    def pending_classes(self) -> List[TaskClass]:
        """Group pending tasks into batching classes.

        Order: priority desc, then role/signature.  Tasks keep dict
        insertion order inside a class (no per-task sort — hot path)."""
        groups: Dict[str, TaskClass] = {}
        for t in self.task_status_index.get(TaskStatus.PENDING, {}).values():
            if t.gated:
                continue    # scheduling gates hold the pod back (k8s gate)
            sig = t.class_signature()
            g = groups.get(sig)
            if g is None:
                g = groups[sig] = TaskClass(signature=sig, role=t.role,
                                            request=t.request,
                                            priority=t.priority)
            g.tasks.append(t)
        if len(groups) == 1:
            return list(groups.values())
        return sorted(groups.values(), key=lambda g: (-g.priority, g.role, g.signature))

    def total_request(self) -> Resource:
        r = Resource()
        for t in self.tasks.values():
            r.add(t.request)
        return r

    def allocated_resource(self) -> Resource:
        """O(allocated tasks) via the status index, not O(all tasks)."""
        r = Resource()
        idx = self.task_status_index
        for s in ALLOCATED_STATUSES:
            for t in idx.get(s, {}).values():
                r.add(t.request)
        return r

    def clone_shell(self) -> "JobInfo":
        j = JobInfo(self.key, self.podgroup)
        j.job_id = self.job_id
        return j


class NodeInfo:
    """Per-node state (reference api/node_info.go:52-97).

    Usage accounting (used/releasing/pipelined/remote_used) lives in the
    cache-level :class:`~volcano_amd.api.ledger.NodeLedger` once the node
    is adopted into the packed order; the properties below materialize
    read-only ``Resource`` snapshots from the ledger row.  Mutations go
    through the node methods (or ledger bulk ops) only."""

    def __init__(self, node: Node):
        self.node = node
        self.name = node.meta.name
        self.node_id = -1                    # dense index
        self.allocatable = node.allocatable.clone()
        # node-agent oversubscription report (reference node_info.go:83-89:
        # oversold capacity from real utilization extends allocatable)
        self.oversubscription = node.oversubscription.clone()
        ann = node.meta.annotations
        if "volcano.sh/oversubscription-cpu" in ann:
            from .resource import CPU
            self.oversubscription.q[CPU] = float(
                ann["volcano.sh/oversubscription-cpu"])
        if "volcano.sh/oversubscription-memory" in ann:
            from .resource import MEMORY
            self.oversubscription.q[MEMORY] = float(
                ann["volcano.sh/oversubscription-memory"])
        self.allocatable.add(self.oversubscription)
        # DRA: device-class capacity advertised per node (the
        # ResourceSlice analog) — becomes the dense dim dra:<class>
        for k, v in ann.items():
            if k.startswith("dra.volcano.sh/"):
                self.allocatable.q[f"dra:{k[len('dra.volcano.sh/'):]}"] = \
                    float(v)
        # pre-adoption local truth (ledger adoption copies these in)
        self._used = Resource()
        self._releasing = Resource()
        self._pipelined = Resource()
        self._remote_used = Resource()
        self._ledger = None
        self._row = -1
        self._tasks: Dict[str, TaskInfo] = {}
        # deferred membership batches (allocate commit appends task lists;
        # the `tasks` property folds them on first read — cold paths only)
        self._batches: List[List[TaskInfo]] = []

    # -- usage views ---------------------------------------------------------
    @property
    def tasks(self) -> Dict[str, TaskInfo]:
        b = self._batches
        if b:
            t = self._tasks
            for ts in b:
                for x in ts:
                    t[x.key] = x
            b.clear()
        return self._tasks

    def _plane(self, plane: int, local: Resource) -> Resource:
        led = self._ledger
        if led is None or self._row < 0:
            return local
        return led.resource(self._row, plane)

    @property
    def used(self) -> Resource:
        from .ledger import USED
        return self._plane(USED, self._used)

    @used.setter
    def used(self, r: Resource) -> None:
        from .ledger import USED
        if self._ledger is None or self._row < 0:
            self._used = r
        else:
            self._ledger.set_row(self._row, USED, r)

    @property
    def releasing(self) -> Resource:
        from .ledger import RELEASING
        return self._plane(RELEASING, self._releasing)

    @releasing.setter
    def releasing(self, r: Resource) -> None:
        from .ledger import RELEASING
        if self._ledger is None or self._row < 0:
            self._releasing = r
        else:
            self._ledger.set_row(self._row, RELEASING, r)

    @property
    def pipelined(self) -> Resource:
        from .ledger import PIPELINED
        return self._plane(PIPELINED, self._pipelined)

    @pipelined.setter
    def pipelined(self, r: Resource) -> None:
        from .ledger import PIPELINED
        if self._ledger is None or self._row < 0:
            self._pipelined = r
        else:
            self._ledger.set_row(self._row, PIPELINED, r)

    @property
    def remote_used(self) -> Resource:
        from .ledger import REMOTE
        return self._plane(REMOTE, self._remote_used)

    @remote_used.setter
    def remote_used(self, r: Resource) -> None:
        from .ledger import REMOTE
        if self._ledger is None or self._row < 0:
            self._remote_used = r
        else:
            self._ledger.set_row(self._row, REMOTE, r)

    @property
    def idle(self) -> Resource:
        led = self._ledger
        if led is None or self._row < 0:
            return self.allocatable.clone().sub(self._used)
        from .ledger import USED
        row = led.alloc[self._row] - led.planes[USED, self._row]
        names = led.dims.names
        out = Resource({names[i]: float(v) for i, v in enumerate(row)
                        if v > 0.0 and i < len(names)})
        return out

    @property
    def future_idle(self) -> Resource:
        """idle + releasing − pipelined (reference FutureIdle)."""
        led = self._ledger
        if led is None or self._row < 0:
            return self.allocatable.clone().sub(self._used) \
                .add(self._releasing).sub(self._pipelined)
        from .ledger import PIPELINED, RELEASING, USED
        p = led.planes
        row = (led.alloc[self._row] - p[USED, self._row]
               + p[RELEASING, self._row] - p[PIPELINED, self._row])
        names = led.dims.names
        return Resource({names[i]: float(v) for i, v in enumerate(row)
                         if v > 0.0 and i < len(names)})

    @property
    def ready(self) -> bool:
        return self.node.ready and not self.node.unschedulable

    # -- accounting mutations -------------------------------------------------
    def _acct(self, request: Resource, du: float, dr: float,
              dp: float) -> None:
        led = self._ledger
        if led is not None and self._row >= 0:
            led.apply(self._row, request.q, du, dr, dp)
            return
        if du > 0:
            self._used.add(request)
        elif du < 0:
            self._used.sub(request)
        if dr > 0:
            self._releasing.add(request)
        elif dr < 0:
            self._releasing.sub(request)
        if dp > 0:
            self._pipelined.add(request)
        elif dp < 0:
            self._pipelined.sub(request)

    def add_task(self, task: TaskInfo) -> None:
        tasks = self.tasks
        prev = tasks.get(task.key)
        if prev is not None and prev is not task:
            self.remove_task(prev)     # never silently overwrite accounting
        tasks[task.key] = task
        if task.status.occupies_node:
            self._acct(task.request, 1, 0, 0)
        elif task.status == TaskStatus.RELEASING:
            self._acct(task.request, 1, 1, 0)
        elif task.status == TaskStatus.PIPELINED:
            self._acct(task.request, 0, 0, 1)

    def add_allocated_bulk(self, tasks: List[TaskInfo], request: Resource,
                           count: int) -> None:
        """Bulk add for freshly-allocated identical tasks: ONE accounting op
        for the whole batch (hot path of the allocate commit)."""
        self._batches.append(tasks)
        led = self._ledger
        if led is not None and self._row >= 0:
            led.apply(self._row, request.q, float(count), 0, 0)
        else:
            self._used.add(request.clone().multi(float(count)))

    def remove_task(self, task: TaskInfo) -> None:
        tasks = self.tasks
        if task.key not in tasks:
            return
        del tasks[task.key]
        if task.status.occupies_node:
            self._acct(task.request, -1, 0, 0)
        elif task.status == TaskStatus.RELEASING:
            self._acct(task.request, -1, -1, 0)
        elif task.status == TaskStatus.PIPELINED:
            self._acct(task.request, 0, 0, -1)


class QueueInfo:
    """Per-queue state (reference api/queue_info.go)."""

    def __init__(self, queue: Queue):
        self.queue = queue
        self.name = queue.meta.name
        self.queue_id = -1

    @property
    def weight(self) -> int:
        return max(int(self.queue.spec.weight), 1)

    @property
    def hierarchy(self) -> str:
        """Slash path in the fair-share tree (reference queue_info.go:48,
        annotation ``volcano.sh/hierarchy``, e.g. ``root/eng/dev``)."""
        return self.queue.meta.annotations.get("volcano.sh/hierarchy", "")

    @property
    def hierarchy_weights(self) -> str:
        """Slash weights along :attr:`hierarchy` (queue_info.go:45)."""
        return self.queue.meta.annotations.get(
            "volcano.sh/hierarchy-weights", "")

    @property
    def capability(self) -> Resource:
        from .resource import normalize_dra_keys
        return normalize_dra_keys(self.queue.spec.capability)

    @property
    def guarantee(self) -> Resource:
        from .resource import normalize_dra_keys
        return normalize_dra_keys(self.queue.spec.guarantee)

    @property
    def deserved_spec(self) -> Resource:
        from .resource import normalize_dra_keys
        return normalize_dra_keys(self.queue.spec.deserved)

    @property
    def reclaimable(self) -> bool:
        return self.queue.spec.reclaimable

    @property
    def parent(self) -> str:
        return self.queue.spec.parent

    @property
    def priority(self) -> int:
        return self.queue.spec.priority

    @property
    def is_open(self) -> bool:
        return self.queue.status.state == "Open"
