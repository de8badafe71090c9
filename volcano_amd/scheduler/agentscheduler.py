"""Agent scheduler — the fast path for non-gang, latency-sensitive pods.

Reference: ``pkg/agentscheduler/`` (design docs/design/agent-scheduler.md):
a separate scheduler with N optimistic-parallel workers, each holding its
own snapshot, running ONLY the allocate action with predicates/nodeorder —
no gang machinery, no queue fairness; cooperates with the main scheduler
through NodeShard hard sharding.

MI355X shape: a pod-at-a-time decision is one ``score_cap`` pass over the
node planes + argmax — single kernel on GPU, one fused torch op on CPU.
Workers are threads pulling from a shared queue; conflicts cannot happen
because each worker commits through the shared cache under the pod lock
(optimistic retry on capacity mismatch, scheduler.go:135 analog).
"""

from __future__ import annotations

import queue
import threading
from typing import Optional

import torch

from ..api.info import NodeInfo, TaskInfo
from ..api.objects import Pod
from ..ops import reference as ref
from ..store import EventType, ObjectStore
from .cache import Binder, SchedulerCache

FAST_PATH_ANN = "volcano.sh/fast-path"


class AgentScheduler:
    def __init__(self, store: ObjectStore, binder: Optional[Binder] = None,
                 workers: int = 4, device: str = "cpu"):
        self.cache = SchedulerCache(store=store, binder=binder, device=device)
        self.workers = workers
        self._q: "queue.Queue[str]" = queue.Queue()
        self._lock = threading.Lock()
        self._watch = store.watch("Pod")
        self.store = store

    def _eligible(self, pod: Pod) -> bool:
        if pod.node_name or pod.phase != "Pending":
            return False
        # fast path: explicitly marked, or a bare pod with no gang group
        return pod.meta.annotations.get(FAST_PATH_ANN) == "true" or \
            not pod.podgroup_name

    def pump(self) -> int:
        """Ingest store events; enqueue eligible pods. Returns queued."""
        n = 0
        for ev in self._watch.drain():
            if ev.type in (EventType.ADDED, EventType.MODIFIED) and \
                    self._eligible(ev.obj):
                self._q.put(ev.obj.meta.key)
                n += 1
        self.cache.sync()
        return n

    def _score_node(self, task: TaskInfo) -> Optional[NodeInfo]:
        """One fused feasibility+score pass, best node wins."""
        nt = self.cache.node_tensors
        nodes = sorted(self.cache.nodes.values(), key=lambda x: x.name)
        if nt.alloc_t is None or not nodes:
            return None
        req = nt.req_vector(task)
        if req is None:
            return None
        N, R = nt.n, nt.r
        dev = nt.alloc_t.device
        tol = nt.tolerated_mask(task.pod.tolerations if task.pod else [])
        require, forbid = nt.selector_bits(
            task.pod.node_selector if task.pod else {},
            task.pod.affinity if task.pod else None)
        nt.ensure_plane_width()
        score = torch.empty(N, device=dev)
        cap = torch.empty(N, dtype=torch.int32, device=dev)
        ref.score_cap(nt.alloc_t.t(), nt.used_t.t(),
                      torch.zeros_like(nt.extra_t.t()), nt.ready.bool(),
                      nt.taint_mask, nt.planes_t.t(),
                      torch.from_numpy(req).to(dev), tol,
                      torch.from_numpy(require).to(dev),
                      torch.from_numpy(forbid).to(dev),
                      1.0, 0.0, 1.0, torch.ones(R, device=dev), None,
                      score, cap)
        best = int(torch.argmax(score))
        if score[best].item() == float("-inf"):
            return None
        return nodes[best]

    def _schedule_one(self, pod_key: str) -> bool:
        ns, name = pod_key.split("/", 1)
        pod = self.store.get("Pod", ns, name)
        if pod is None or pod.node_name:
            return False
        with self._lock:     # commit section: capacity check + bind
            self.cache.sync()
            self.cache.ensure_packed()
            job_key = f"{ns}/{pod.podgroup_name or 'pod-' + name}"
            task = None
            job = self.cache.jobs.get(job_key)
            if job is not None:
                task = job.tasks.get(pod_key)
            if task is None:
                task = TaskInfo.from_pod(pod, job_key)
            ni = self._score_node(task)
            if ni is None:
                return False
            task.node_name = ni.name
            ni.add_allocated_bulk([task], task.request, 1)
            self.cache._used_dirty = True
            self.cache.bind_tasks([task])
        return True

    def run_workers(self, max_pods: Optional[int] = None) -> int:
        """Drain the queue with N worker threads; returns bound count."""
        done = []

        def worker():
            while True:
                try:
                    key = self._q.get_nowait()
                except queue.Empty:
                    return
                if self._schedule_one(key):
                    done.append(key)

        threads = [threading.Thread(target=worker)
                   for _ in range(self.workers)]
        for t in threads:
            t.start()
        for t in threads:
            t.join()
        return len(done)

    def run_once(self) -> int:
        self.pump()
        return self.run_workers()
