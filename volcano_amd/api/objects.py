"""CRD-style API objects — the framework's "protocol" layer.

Python analogs of the reference's API groups
(``staging/src/volcano.sh/apis/pkg/apis/``):

* batch/v1alpha1  — Job (``job.go:55-130``), CronJob
* scheduling/v1beta1 — PodGroup, Queue (``types.go:174-240, 461-508``)
* bus/v1alpha1    — Command (``commands.go:13-28``)
* topology/v1alpha1 — HyperNode (``hypernode_types.go:61-106``)
* shard/v1alpha1  — NodeShard (``types.go:33-71``)
* flow/v1alpha1   — JobFlow / JobTemplate (``jobflow_types.go:26-41``)

plus the core kinds the scheduler consumes (Node, Pod).  Objects are plain
dataclasses with dict/YAML round-trip; the in-process object store
(``volcano_amd/store``) plays the role of kube-apiserver/etcd.
"""

from __future__ import annotations

import dataclasses
import time
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

from .resource import Resource
from .types import JobPhase, PodGroupPhase, QueueState

DEFAULT_NAMESPACE = "default"
DEFAULT_QUEUE = "default"
DEFAULT_SCHEDULER = "volcano"

# Annotation / label keys (mirror the reference's well-known keys).
ANN_QUEUE = "scheduling.volcano.sh/queue-name"
ANN_PODGROUP = "scheduling.volcano.sh/group-name"
ANN_PREEMPTABLE = "volcano.sh/preemptable"
LBL_JOB_NAME = "volcano.sh/job-name"
LBL_TASK_SPEC = "volcano.sh/task-spec"
LBL_TASK_INDEX = "volcano.sh/task-index"
LBL_NODEGROUP = "volcano.sh/nodegroup-name"
LBL_REVOCABLE_ZONE = "volcano.sh/revocable-zone"


def _ts() -> float:
    return time.time()


@dataclass
class ObjectMeta:
    name: str = ""
    namespace: str = DEFAULT_NAMESPACE
    labels: Dict[str, str] = field(default_factory=dict)
    annotations: Dict[str, str] = field(default_factory=dict)
    uid: str = ""
    resource_version: int = 0
    creation_timestamp: float = field(default_factory=_ts)
    owner: Optional[str] = None      # "Kind/namespace/name" back-reference
    deletion_timestamp: Optional[float] = None

    @property
    def key(self) -> str:
        return f"{self.namespace}/{self.name}"


@dataclass
class Toleration:
    key: str = ""
    operator: str = "Equal"      # Equal | Exists
    value: str = ""
    effect: str = ""             # "", NoSchedule, PreferNoSchedule, NoExecute

    def tolerates(self, taint: "Taint") -> bool:
        if self.effect and self.effect != taint.effect:
            return False
        if self.operator == "Exists":
            return not self.key or self.key == taint.key
        return self.key == taint.key and self.value == taint.value


@dataclass
class Taint:
    key: str = ""
    value: str = ""
    effect: str = "NoSchedule"


@dataclass
class Node:
    """core/v1 Node as the scheduler sees it."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    allocatable: Resource = field(default_factory=Resource)
    capacity: Resource = field(default_factory=Resource)
    taints: List[Taint] = field(default_factory=list)
    unschedulable: bool = False
    ready: bool = True
    # colocation / oversubscription (reference node_info.go:83-89)
    oversubscription: Resource = field(default_factory=Resource)
    images: List[str] = field(default_factory=list)  # cached container images


@dataclass
class Namespace:
    """core/v1 Namespace — carried for the queue-defaulting admission
    path (reference mutate_podgroup.go:104-125: a namespace may pin its
    workloads to a queue via the ``scheduling.volcano.sh/queue-name``
    annotation)."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    phase: str = "Active"


@dataclass
class Pod:
    """core/v1 Pod reduced to scheduling-relevant fields."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    request: Resource = field(default_factory=Resource)
    node_name: str = ""                 # bound node ("" = unscheduled)
    phase: str = "Pending"              # Pending/Running/Succeeded/Failed
    priority: int = 0
    priority_class: str = ""
    scheduler_name: str = DEFAULT_SCHEDULER
    tolerations: List[Toleration] = field(default_factory=list)
    node_selector: Dict[str, str] = field(default_factory=dict)
    affinity: Optional[dict] = None     # simplified nodeAffinity expression tree
    best_effort: bool = field(default=False)
    scheduling_gates: List[str] = field(default_factory=list)
    volumes: List[str] = field(default_factory=list)   # PVC names (same ns)
    # k8s pod.spec.preemptionPolicy: "" (PreemptLowerPriority) | "Never" —
    # a Never preemptor may wait but must not displace anything
    # (reference preempt.go:441 taskEligibleToPreempt, #3642)
    preemption_policy: str = ""
    host_ports: List[int] = field(default_factory=list)  # requested hostPorts
    image: str = ""                                    # container image
    # DRA (k8s dynamic resource allocation): ResourceClaim names (same
    # namespace) the pod consumes — each claim names a DeviceClass
    resource_claims: List[str] = field(default_factory=list)

    def __post_init__(self):
        # BestEffort = no resource requests (reference pod_info.go); the
        # implicit one-pod slot every pod occupies doesn't count
        if not any(v > 1e-9 for k, v in self.request.q.items()
                   if k != "pods"):
            self.best_effort = True

    @property
    def podgroup_name(self) -> str:
        return self.meta.annotations.get(ANN_PODGROUP, "")

    @property
    def queue_name(self) -> str:
        return self.meta.annotations.get(ANN_QUEUE, DEFAULT_QUEUE)


# ---------------------------------------------------------------------------
# batch.volcano.sh/v1alpha1 Job
# ---------------------------------------------------------------------------

@dataclass
class LifecyclePolicy:
    """event → action mapping (job.go policies)."""

    event: str = "*"
    events: List[str] = field(default_factory=list)
    action: str = ""
    exit_code: Optional[int] = None
    timeout_seconds: Optional[float] = None

    def matches(self, event: str, exit_code: Optional[int] = None) -> bool:
        if self.exit_code is not None:
            return exit_code == self.exit_code
        evs = self.events or ([self.event] if self.event else [])
        return event in evs or "*" in evs


@dataclass
class TaskSpec:
    """One task role of a vcjob (job.go TaskSpec)."""

    name: str = ""
    replicas: int = 1
    min_available: Optional[int] = None
    template: Dict[str, Any] = field(default_factory=dict)  # pod template (request, tolerations, ...)
    policies: List[LifecyclePolicy] = field(default_factory=list)
    topology_policy: str = ""        # none/best-effort/restricted/single-numa-node
    depends_on: List[str] = field(default_factory=list)
    max_retry: int = 3

    def request(self) -> Resource:
        return Resource.from_spec(self.template.get("resources", {}))

    @property
    def min_needed(self) -> int:
        return self.replicas if self.min_available is None else self.min_available


@dataclass
class JobSpec:
    scheduler_name: str = DEFAULT_SCHEDULER
    min_available: Optional[int] = None
    volumes: List[dict] = field(default_factory=list)
    tasks: List[TaskSpec] = field(default_factory=list)
    policies: List[LifecyclePolicy] = field(default_factory=list)
    plugins: Dict[str, List[str]] = field(default_factory=dict)
    queue: str = DEFAULT_QUEUE
    max_retry: int = 3
    ttl_seconds_after_finished: Optional[float] = None
    priority_class: str = ""
    min_success: Optional[int] = None
    running_estimate: Optional[float] = None
    network_topology: Optional[dict] = None   # {"mode": "hard"|"soft", "highestTierAllowed": int}

    @property
    def total_replicas(self) -> int:
        return sum(t.replicas for t in self.tasks)

    @property
    def effective_min_available(self) -> int:
        if self.min_available is not None:
            return self.min_available
        return sum(t.min_needed for t in self.tasks)


@dataclass
class JobStatus:
    phase: str = JobPhase.PENDING.value
    pending: int = 0
    running: int = 0
    succeeded: int = 0
    failed: int = 0
    terminating: int = 0
    unknown: int = 0
    version: int = 0
    retry_count: int = 0
    conditions: List[dict] = field(default_factory=list)


@dataclass
class Job:
    """batch.volcano.sh/v1alpha1 Job (job.go:55-130)."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    spec: JobSpec = field(default_factory=JobSpec)
    status: JobStatus = field(default_factory=JobStatus)


# ---------------------------------------------------------------------------
# scheduling.volcano.sh/v1beta1 PodGroup / Queue
# ---------------------------------------------------------------------------

@dataclass
class PodGroupSpec:
    min_member: int = 1
    min_task_member: Dict[str, int] = field(default_factory=dict)
    queue: str = DEFAULT_QUEUE
    priority_class: str = ""
    min_resources: Resource = field(default_factory=Resource)
    network_topology: Optional[dict] = None
    # SubGroupPolicy (reference scheduling/v1beta1 types.go:218-267):
    # secondary grouping — each entry {"name", "subGroupSize",
    # "minSubGroups", "networkTopology", "labelSelector",
    # "matchLabelKeys"} partitions matching pods into gang-atomic
    # subgroups of subGroupSize; the job commits only when at least
    # minSubGroups of them place.
    sub_group_policy: List[dict] = field(default_factory=list)


@dataclass
class PodGroupStatus:
    phase: str = PodGroupPhase.PENDING.value
    conditions: List[dict] = field(default_factory=list)
    running: int = 0
    succeeded: int = 0
    failed: int = 0


@dataclass
class PodGroup:
    meta: ObjectMeta = field(default_factory=ObjectMeta)
    spec: PodGroupSpec = field(default_factory=PodGroupSpec)
    status: PodGroupStatus = field(default_factory=PodGroupStatus)


@dataclass
class QueueSpec:
    weight: int = 1
    capability: Resource = field(default_factory=Resource)
    reclaimable: bool = True
    guarantee: Resource = field(default_factory=Resource)
    deserved: Resource = field(default_factory=Resource)   # capacity plugin
    parent: str = ""                                       # hierarchy
    priority: int = 0
    type: str = ""
    # nodegroup affinity (reference Queue.spec.affinity — plugins/nodegroup):
    # {"affinity": {"required": [...], "preferred": [...]},
    #  "antiAffinity": {"required": [...], "preferred": [...]}}
    affinity: Optional[Dict[str, Any]] = None
    # fifo: a job that cannot enqueue blocks the jobs behind it;
    # traverse (default): keep trying the rest (reference dequeueStrategy)
    dequeue_strategy: str = "traverse"
    # multi-cluster dispatch targets (reference Queue spec extendClusters,
    # types.go:461-508 — incubating; carried for API-surface parity)
    extend_clusters: List[Dict[str, Any]] = field(default_factory=list)


@dataclass
class QueueStatus:
    state: str = QueueState.OPEN.value
    pending: int = 0
    running: int = 0
    inqueue: int = 0
    allocated: Resource = field(default_factory=Resource)


@dataclass
class Queue:
    meta: ObjectMeta = field(default_factory=ObjectMeta)
    spec: QueueSpec = field(default_factory=QueueSpec)
    status: QueueStatus = field(default_factory=QueueStatus)


# ---------------------------------------------------------------------------
# bus / topology / shard / flow groups
# ---------------------------------------------------------------------------

@dataclass
class Command:
    """bus/v1alpha1 Command — async control channel (commands.go:13-28)."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    action: str = ""
    target_kind: str = ""
    target_name: str = ""
    reason: str = ""
    message: str = ""


@dataclass
class MemberSelector:
    exact_match: List[str] = field(default_factory=list)
    regex_match: str = ""
    label_match: Dict[str, str] = field(default_factory=dict)


@dataclass
class HyperNodeMember:
    type: str = "Node"        # Node | HyperNode
    selector: MemberSelector = field(default_factory=MemberSelector)


@dataclass
class HyperNode:
    """topology/v1alpha1 HyperNode — recursive network-domain tree."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    tier: int = 1
    members: List[HyperNodeMember] = field(default_factory=list)


@dataclass
class NodeShard:
    """shard/v1alpha1 NodeShard — 2-phase node handoff between schedulers."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    nodes_desired: List[str] = field(default_factory=list)
    nodes_in_use: List[str] = field(default_factory=list)
    nodes_to_add: List[str] = field(default_factory=list)
    nodes_to_remove: List[str] = field(default_factory=list)


ZONE_LABEL = "topology.kubernetes.io/zone"


@dataclass
class PersistentVolume:
    """core/v1 PV reduced to the volume-zone filter's needs: a bound PV
    whose zone label confines its consumers' pods to that zone (k8s
    volumezone filter, wrapped by reference plugins/predicates)."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    capacity: float = 0.0               # bytes
    storage_class: str = ""
    claim_ref: str = ""                 # "ns/name" of the bound PVC ("" = free)
    # zone read from meta.labels[ZONE_LABEL] ("" = zone-free volume)

    @property
    def zone(self) -> str:
        return self.meta.labels.get(ZONE_LABEL, "")


@dataclass
class PersistentVolumeClaim:
    """core/v1 PVC: binds a pod's volume to a PV.  Unbound claims
    (volume_name == "") constrain nothing — WaitForFirstConsumer
    semantics: the volume binds where the pod lands."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    volume_name: str = ""               # bound PV ("" = unbound)
    storage_class: str = ""
    request: float = 0.0                # bytes


@dataclass
class ResourceQuota:
    """core/v1 ResourceQuota as the resourcequota plugin consumes it:
    per-namespace hard limits checked at job enqueue."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    hard: Resource = field(default_factory=Resource)
    used: Resource = field(default_factory=Resource)


@dataclass
class ColocationConfig:
    """config/v1alpha1 ColocationConfig — per-nodepool colocation/QoS
    settings (cpu burst, memory qos, oversubscription ratio, network
    bandwidth watermarks) keyed by a node label selector."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    node_selector: Dict[str, str] = field(default_factory=dict)
    cpu_burst_enable: bool = False
    memory_qos_enable: bool = False
    oversubscription_enable: bool = False
    oversubscription_ratio: float = 0.6
    network_qos_enable: bool = False
    offline_bandwidth_share: float = 0.3


@dataclass
class NumaZone:
    id: int = 0
    cpu_milli: float = 0.0       # allocatable millicores in this NUMA node
    memory: float = 0.0
    # explicit CPU id list for cpuset pinning (reference numaaware
    # providers); empty → derived as a contiguous range from cpu_milli
    cpus: List[int] = field(default_factory=list)


@dataclass
class Numatopology:
    """nodeinfo/v1alpha1 Numatopology — per-node CPU/NUMA layout written
    by the node agent, consumed by the numaaware plugin."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)   # name == node name
    zones: List[NumaZone] = field(default_factory=list)
    policies: Dict[str, str] = field(default_factory=dict)  # e.g. topologyManager


@dataclass
class PodDisruptionBudget:
    """policy/v1 PodDisruptionBudget as the pdb plugin consumes it."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    selector: Dict[str, str] = field(default_factory=dict)   # label match
    min_available: Optional[int] = None
    max_unavailable: Optional[int] = None


@dataclass
class FlowStep:
    name: str = ""
    depends_on: List[str] = field(default_factory=list)
    patch: Dict[str, Any] = field(default_factory=dict)


@dataclass
class JobFlow:
    """flow/v1alpha1 JobFlow — DAG of JobTemplates."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    flows: List[FlowStep] = field(default_factory=list)
    job_retain_policy: str = "retain"    # retain | delete
    status: Dict[str, Any] = field(default_factory=dict)


@dataclass
class JobTemplate:
    meta: ObjectMeta = field(default_factory=ObjectMeta)
    spec: JobSpec = field(default_factory=JobSpec)
    # reference jobtemplate controller bookkeeping: names of Jobs
    # created from this template (status.jobDependsOnList)
    status: Dict[str, Any] = field(default_factory=dict)


@dataclass
class HyperJob:
    """training/v1alpha1 HyperJob (reference apis/training — incubating):
    one logical job split into per-replica member Jobs (the multi-cluster
    splitting shape, applied here to replica groups)."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    replicas: int = 1                       # member jobs to create
    job_template: JobSpec = field(default_factory=JobSpec)
    status: Dict[str, Any] = field(default_factory=dict)


@dataclass
class DeviceClass:
    """resource.k8s.io DeviceClass (DRA): a named class of devices.
    Scheduling rides the dense dims — a class becomes the synthetic
    resource dim ``dra:<name>``; nodes advertise capacity via the
    ``dra.volcano.sh/<name>`` annotation (the ResourceSlice analog).
    Reference: plugins/predicates dynamicresources wrap
    (predicates.go:34-47) + capacity DRA quotas (capacity.go:107-197)."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    driver: str = ""                      # informational (device plugin)
    config: Dict[str, Any] = field(default_factory=dict)


@dataclass
class ResourceClaim:
    """resource.k8s.io ResourceClaim (DRA): a pod's claim for N devices
    of a DeviceClass."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    device_class_name: str = ""
    count: int = 1
    # filled at bind: node the devices were allocated on
    allocated_node: str = ""


@dataclass
class DataSource:
    """datadependency/v1alpha1 DataSource: a cached, cluster-scoped
    record of where a logical data source (hive table, s3 bucket, hdfs
    path ...) is available.  Reference:
    staging/.../datadependency/v1alpha1/types.go:32-103."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    system: str = ""                    # "hive" | "s3" | "hdfs" | ...
    type: str = ""                      # "table" | "bucket" | ...
    name: str = ""                      # logical name within the system
    cluster_names: List[str] = field(default_factory=list)  # locality
    attributes: Dict[str, str] = field(default_factory=dict)
    reclaim_policy: str = "Retain"      # Retain | Delete
    # status
    claim_refs: List[str] = field(default_factory=list)   # bound claims
    bound_claims: int = 0


@dataclass
class DataSourceClaim:
    """datadependency/v1alpha1 DataSourceClaim: a workload's request for
    a DataSource, matched by (system, dataSourceType, dataSourceName).
    Reference: staging/.../datadependency/v1alpha1/types.go:124-201."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    system: str = ""
    data_source_type: str = ""
    data_source_name: str = ""
    workload: Dict[str, str] = field(default_factory=dict)  # WorkloadRef
    attributes: Dict[str, str] = field(default_factory=dict)
    # status
    phase: str = "Pending"              # Pending | Bound | Lost
    bound_data_source: str = ""


@dataclass
class CronJob:
    """batch/v1alpha1 CronJob (reference cronjob controller)."""

    meta: ObjectMeta = field(default_factory=ObjectMeta)
    schedule: str = "* * * * *"
    concurrency_policy: str = "Allow"    # Allow | Forbid | Replace
    starting_deadline_seconds: Optional[float] = None
    successful_jobs_history_limit: int = 3
    failed_jobs_history_limit: int = 1
    suspend: bool = False
    job_template: JobSpec = field(default_factory=JobSpec)
    status: Dict[str, Any] = field(default_factory=dict)


# ---------------------------------------------------------------------------
# dict / yaml round-trip
# ---------------------------------------------------------------------------

def to_dict(obj) -> Any:
    if dataclasses.is_dataclass(obj) and not isinstance(obj, type):
        return {f.name: to_dict(getattr(obj, f.name)) for f in dataclasses.fields(obj)}
    if isinstance(obj, Resource):
        return dict(obj.q)
    if isinstance(obj, dict):
        return {k: to_dict(v) for k, v in obj.items()}
    if isinstance(obj, (list, tuple)):
        return [to_dict(v) for v in obj]
    return obj


def from_dict(cls, data: Any):
    """Reconstruct a dataclass tree from its ``to_dict`` output."""
    import typing

    if data is None:
        return None
    if cls is Resource:
        return Resource(data)
    origin = typing.get_origin(cls)
    if origin in (list, tuple):
        (elt,) = typing.get_args(cls) or (Any,)
        return [from_dict(elt, v) for v in data]
    if origin is dict:
        return dict(data)
    if origin is typing.Union:  # Optional[X]
        args = [a for a in typing.get_args(cls) if a is not type(None)]
        return from_dict(args[0], data) if len(args) == 1 else data
    if isinstance(cls, type) and dataclasses.is_dataclass(cls):
        hints = typing.get_type_hints(cls)
        kwargs = {}
        for f in dataclasses.fields(cls):
            if f.name in data:
                kwargs[f.name] = from_dict(hints.get(f.name, Any), data[f.name])
        return cls(**kwargs)
    return data


# kind registry for the object store / CLI
KINDS = {
    "Node": Node, "Pod": Pod, "Job": Job, "PodGroup": PodGroup,
    "Queue": Queue, "Command": Command, "HyperNode": HyperNode,
    "NodeShard": NodeShard, "JobFlow": JobFlow, "JobTemplate": JobTemplate,
    "CronJob": CronJob, "PodDisruptionBudget": PodDisruptionBudget,
    "Numatopology": Numatopology, "ColocationConfig": ColocationConfig,
    "ResourceQuota": ResourceQuota, "HyperJob": HyperJob,
    "PersistentVolume": PersistentVolume,
    "PersistentVolumeClaim": PersistentVolumeClaim,
    "DeviceClass": DeviceClass, "ResourceClaim": ResourceClaim,
    "DataSource": DataSource, "DataSourceClaim": DataSourceClaim,
    "Namespace": Namespace,
}
