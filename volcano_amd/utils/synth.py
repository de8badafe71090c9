"""Synthetic inventory builders — the KWOK-analog test/bench substrate.

The reference measures scheduling performance on KWOK-simulated clusters
(benchmark/README.md: virtual nodes, e.g. 100×32CPU/256Gi; the 10k-pods
study of docs/user-guide/how_to_tune_volcano_performance.md).  These
helpers build the same shaped inventories as API objects (for store-driven
runs) or directly as infos (for kernel-level tests).
"""

from __future__ import annotations

import random
from typing import Dict, List, Optional

from ..api.objects import (ANN_PODGROUP, ANN_QUEUE, LBL_JOB_NAME,
                           LBL_TASK_SPEC, Node,
                           ObjectMeta, Pod, PodGroup, PodGroupSpec, Queue,
                           QueueSpec, Taint, Toleration)
from ..api.resource import CPU, MEMORY, PODS, Resource


def make_node(name: str, cpu_milli: float = 32000,
              mem: float = 256 * 1024 ** 3, pods: int = 110,
              labels: Optional[Dict[str, str]] = None,
              taints: Optional[List[Taint]] = None,
              extra: Optional[Dict[str, float]] = None) -> Node:
    alloc = {CPU: cpu_milli, MEMORY: mem, PODS: float(pods)}
    if extra:
        alloc.update(extra)
    return Node(meta=ObjectMeta(name=name, labels=labels or {}),
                allocatable=Resource(alloc), capacity=Resource(alloc),
                taints=taints or [])


def make_nodes(n: int, prefix: str = "node", **kw) -> List[Node]:
    return [make_node(f"{prefix}-{i:05d}", **kw) for i in range(n)]


def make_queue(name: str, weight: int = 1,
               capability: Optional[Dict[str, float]] = None,
               guarantee: Optional[Dict[str, float]] = None,
               reclaimable: bool = True,
               parent: str = "",
               deserved: Optional[Dict[str, float]] = None,
               priority: int = 0,
               state: Optional[str] = None) -> Queue:
    q = Queue(meta=ObjectMeta(name=name, namespace="default"),
              spec=QueueSpec(weight=weight,
                             capability=Resource(capability or {}),
                             guarantee=Resource(guarantee or {}),
                             reclaimable=reclaimable,
                             parent=parent,
                             priority=priority,
                             deserved=Resource(deserved or {})))
    if state is not None:
        q.status.state = state
    return q


def make_podgroup(name: str, queue: str = "default", min_member: int = 1,
                  namespace: str = "default",
                  min_task_member: Optional[Dict[str, int]] = None,
                  min_resources: Optional[Dict[str, float]] = None) -> PodGroup:
    return PodGroup(
        meta=ObjectMeta(name=name, namespace=namespace),
        spec=PodGroupSpec(min_member=min_member, queue=queue,
                          min_task_member=min_task_member or {},
                          min_resources=Resource(min_resources or {})))


def make_pod(name: str, podgroup: str, queue: str = "default",
             namespace: str = "default", cpu_milli: float = 1000,
             mem: float = 1024 ** 3, role: str = "",
             priority: int = 0, node_name: str = "",
             node_selector: Optional[Dict[str, str]] = None,
             tolerations: Optional[List[Toleration]] = None,
             extra: Optional[Dict[str, float]] = None,
             phase: str = "Pending") -> Pod:
    req = {}
    if cpu_milli:
        req[CPU] = cpu_milli
    if mem:
        req[MEMORY] = mem
    req[PODS] = 1.0
    if extra:
        req.update(extra)
    labels = {LBL_TASK_SPEC: role} if role else {}
    if podgroup:
        labels[LBL_JOB_NAME] = podgroup
    return Pod(
        meta=ObjectMeta(name=name, namespace=namespace, labels=labels,
                        annotations={ANN_PODGROUP: podgroup, ANN_QUEUE: queue}),
        request=Resource(req), node_name=node_name, priority=priority,
        node_selector=node_selector or {}, tolerations=tolerations or [],
        phase=phase)


def make_gang(store, name: str, replicas: int, queue: str = "default",
              cpu_milli: float = 1000, mem: float = 1024 ** 3,
              min_member: Optional[int] = None, role: str = "worker",
              namespace: str = "default", priority: int = 0,
              min_task_member: Optional[Dict[str, int]] = None,
              **pod_kw) -> PodGroup:
    """One PodGroup + its pods into the store (vcjob-shaped)."""
    mm = replicas if min_member is None else min_member
    pg = make_podgroup(name, queue=queue, min_member=mm, namespace=namespace,
                       min_task_member=min_task_member,
                       min_resources={CPU: cpu_milli * mm, MEMORY: mem * mm})
    store.create("PodGroup", pg)
    for i in range(replicas):
        store.create("Pod", make_pod(
            f"{name}-{role}-{i}", podgroup=name, queue=queue,
            namespace=namespace, cpu_milli=cpu_milli, mem=mem, role=role,
            priority=priority, **pod_kw))
    return pg


def populate(store, n_nodes: int = 100, n_jobs: int = 10,
             pods_per_job: int = 5, queues: Optional[List[str]] = None,
             node_cpu: float = 32000, node_mem: float = 256 * 1024 ** 3,
             pod_cpu: float = 1000, pod_mem: float = 1024 ** 3,
             seed: int = 0) -> None:
    """A whole synthetic cluster in the store."""
    rng = random.Random(seed)
    for node in make_nodes(n_nodes, cpu_milli=node_cpu, mem=node_mem):
        store.create("Node", node)
    qs = queues or ["default"]
    for q in qs:
        if store.get("Queue", "default", q) is None:
            store.create("Queue", make_queue(q))
    for j in range(n_jobs):
        make_gang(store, f"job-{j:05d}", replicas=pods_per_job,
                  queue=qs[j % len(qs)], cpu_milli=pod_cpu, mem=pod_mem)
