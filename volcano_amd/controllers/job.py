"""Job controller — the vcjob state machine.

Reference: ``pkg/controllers/job/`` (job_controller_actions.go syncJob
:348-630, createOrUpdatePodGroup :796, state machine job/state/*.go,
lifecycle policies bus/v1alpha1).  One sync reconciles a Job: ensure its
PodGroup exists (minMember/minResources), create/delete pods to match the
task specs, update status counters, and run the phase state machine;
lifecycle events (PodFailed, TaskCompleted, Command) map to actions
(RestartJob/AbortJob/CompleteJob/... bounded by maxRetry).
"""

from __future__ import annotations

from typing import Dict, List, Optional

from ..api.objects import (ANN_PODGROUP, ANN_QUEUE, LBL_JOB_NAME,
                           LBL_TASK_INDEX, LBL_TASK_SPEC, Job, ObjectMeta,
                           Pod, PodGroup, PodGroupSpec, Toleration)
from ..api.resource import Resource
from ..api.types import Action, Event, JobPhase, PodGroupPhase
from ..store import EventType
from .framework import Controller, register
from .jobplugins import JOB_PLUGINS

FINAL_PHASES = (JobPhase.COMPLETED.value, JobPhase.FAILED.value,
                JobPhase.TERMINATED.value, JobPhase.ABORTED.value)


@register("job")
class JobController(Controller):
    watch_kinds = ("Job", "Pod", "Command")

    #: sharded work queues (reference job_controller.go:139-186,313-325:
    #: a job key is hashed to ONE of N workers so per-job ordering holds
    #: while distinct jobs reconcile concurrently)
    workers = 4
    #: error-resync backoff (reference job_controller.go:145 errTasks
    #: rate-limited queue): base delay, doubled per consecutive failure
    err_backoff_base = 0.05
    err_backoff_max = 5.0

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty: set = set()
        # level-triggered dedup: store events reference live objects, so a
        # queued event shows the pod's LATEST state at drain time — fire
        # lifecycle policies only on an observed phase *transition*
        self._pod_phase_seen: Dict[str, str] = {}
        # key → (retry deadline, consecutive failures)
        self._err_queue: Dict[str, tuple] = {}

    @staticmethod
    def _shard_of(key: str, workers: int) -> int:
        import zlib
        return zlib.crc32(key.encode()) % max(workers, 1)

    # -- event routing --------------------------------------------------------
    def handle(self, ev) -> None:
        from ..utils.features import enabled
        if not enabled("VolcanoJobSupport"):
            return
        if ev.kind == "Job":
            if ev.type != EventType.DELETED:
                self._dirty.add(ev.obj.meta.key)
            else:
                self._cleanup_job(ev.obj)
        elif ev.kind == "Pod":
            job_name = ev.obj.meta.labels.get(LBL_JOB_NAME)
            if job_name:
                key = f"{ev.obj.meta.namespace}/{job_name}"
                self._dirty.add(key)
                pkey = ev.obj.meta.key
                if ev.type == EventType.DELETED:
                    self._pod_phase_seen.pop(pkey, None)
                elif self._pod_phase_seen.get(pkey) != ev.obj.phase:
                    self._pod_phase_seen[pkey] = ev.obj.phase
                    if ev.type == EventType.MODIFIED:
                        self._pod_event(key, ev.obj)
        elif ev.kind == "Command":
            self._command(ev)

    def resync(self) -> None:
        import time as _time
        dirty, self._dirty = self._dirty, set()
        # retryable failures whose backoff deadline passed re-enter the
        # worksheet (errTasks resync)
        now = _time.monotonic()
        for key, (deadline, _) in list(self._err_queue.items()):
            if deadline <= now:
                dirty.add(key)
        if not dirty:
            return
        # shard by key hash — per-job ordering inside a shard, shards
        # reconcile concurrently (the reference's N worker goroutines)
        shards: List[List[str]] = [[] for _ in range(self.workers)]
        for key in dirty:
            shards[self._shard_of(key, self.workers)].append(key)

        def run_shard(keys: List[str]) -> None:
            for key in keys:
                ns, name = key.split("/", 1)
                job = self.store.get("Job", ns, name)
                if job is None:
                    self._err_queue.pop(key, None)
                    continue
                try:
                    self.sync_job(job)
                except Exception:
                    _, fails = self._err_queue.get(key, (0.0, 0))
                    fails += 1
                    delay = min(self.err_backoff_base * (2 ** (fails - 1)),
                                self.err_backoff_max)
                    self._err_queue[key] = (_time.monotonic() + delay,
                                            fails)
                else:
                    self._err_queue.pop(key, None)

        live = [sh for sh in shards if sh]
        if len(live) <= 1:
            for sh in live:
                run_shard(sh)
        else:
            import threading
            ts = [threading.Thread(target=run_shard, args=(sh,))
                  for sh in live]
            for t in ts:
                t.start()
            for t in ts:
                t.join()

    # -- lifecycle events → actions (bus/v1alpha1) ----------------------------
    def _pod_event(self, job_key: str, pod: Pod) -> None:
        ns, name = job_key.split("/", 1)
        job = self.store.get("Job", ns, name)
        if job is None or job.status.phase in FINAL_PHASES:
            return
        if pod.phase == "Failed":
            evicted = "volcano.sh/evicted" in pod.meta.annotations
            event = Event.POD_EVICTED.value if evicted else Event.POD_FAILED.value
            task = pod.meta.labels.get(LBL_TASK_SPEC, "")
            self._apply_policies(job, event, task_name=task, pod=pod)

    def _task_policies(self, job: Job, task_name: str):
        for ts in job.spec.tasks:
            if ts.name == task_name:
                return ts.policies
        return []

    def _apply_policies(self, job: Job, event: str, task_name: str = "",
                        pod: Optional[Pod] = None) -> None:
        action = None
        for pol in self._task_policies(job, task_name):
            if pol.matches(event):
                action = pol.action
                break
        if action is None:
            for pol in job.spec.policies:
                if pol.matches(event):
                    action = pol.action
                    break
        if action is None:
            return
        self.execute_action(job, action, task_name=task_name, pod=pod)

    def execute_action(self, job: Job, action: str, task_name: str = "",
                       pod: Optional[Pod] = None) -> None:
        """bus/v1alpha1 actions (actions.go:19-39) → state transitions.
        Per-state action sets follow the reference state machines
        (controllers/job/state/*.go, job_state_test.go): finished jobs
        ignore every action; an Aborted job acts only on Resume (which
        moves it to Restarting with a retry bump — aborted.go:29-40)."""
        phase = job.status.phase
        # Aborted is resumable, so it is NOT in the finished set here
        if phase in FINAL_PHASES and phase != JobPhase.ABORTED.value:
            return                      # finishedState: no-op
        if phase == JobPhase.ABORTED.value:
            if action == Action.RESUME_JOB.value:
                job.status.retry_count += 1
                job.status.phase = JobPhase.RESTARTING.value
                self.store.update("Job", job)
                self._dirty.add(job.meta.key)
            return                      # abortedState: other actions no-op
        if action == Action.ABORT_JOB.value:
            job.status.phase = JobPhase.ABORTING.value
        elif action == Action.TERMINATE_JOB.value:
            job.status.phase = JobPhase.TERMINATING.value
        elif action == Action.COMPLETE_JOB.value:
            job.status.phase = JobPhase.COMPLETING.value
        elif action == Action.RESTART_JOB.value:
            if job.status.retry_count >= job.spec.max_retry:
                job.status.phase = JobPhase.FAILED.value
            else:
                job.status.retry_count += 1
                job.status.phase = JobPhase.RESTARTING.value
        elif action == Action.RESTART_TASK.value and task_name:
            for p in self._job_pods(job):
                if p.meta.labels.get(LBL_TASK_SPEC) == task_name:
                    self.store.delete("Pod", p.meta.namespace, p.meta.name)
        elif action == Action.RESTART_POD.value and pod is not None:
            self.store.delete("Pod", pod.meta.namespace, pod.meta.name)
        elif action == Action.RESUME_JOB.value:
            if job.status.phase in (JobPhase.ABORTED.value,
                                    JobPhase.ABORTING.value):
                job.status.phase = JobPhase.PENDING.value
        self.store.update("Job", job)
        self._dirty.add(job.meta.key)

    def _command(self, ev) -> None:
        if ev.type == EventType.DELETED:
            return
        cmd = ev.obj
        if cmd.target_kind != "Job":
            return
        job = self.store.get("Job", cmd.meta.namespace, cmd.target_name)
        if job is not None:
            self.execute_action(job, cmd.action)
        self.store.delete("Command", cmd.meta.namespace, cmd.meta.name)

    # -- reconcile ------------------------------------------------------------
    def _job_pods(self, job: Job) -> List[Pod]:
        return self.store.list(
            "Pod", namespace=job.meta.namespace,
            selector=lambda p: p.meta.labels.get(LBL_JOB_NAME) == job.meta.name)

    def _pod_name(self, job: Job, task_name: str, index: int) -> str:
        return f"{job.meta.name}-{task_name}-{index}"

    def _min_resources(self, job: Job) -> Resource:
        """PodGroup.minResources: sum of the min-needed task requests
        (reference job_controller_actions.go:932)."""
        total = Resource()
        for ts in job.spec.tasks:
            req = ts.request()
            for _ in range(ts.min_needed):
                total.add(req)
        return total

    def _ensure_podgroup(self, job: Job) -> None:
        pg = self.store.get("PodGroup", job.meta.namespace, job.meta.name)
        if pg is None:
            pg = PodGroup(
                meta=ObjectMeta(name=job.meta.name,
                                namespace=job.meta.namespace,
                                owner=f"Job/{job.meta.key}"),
                spec=PodGroupSpec(
                    min_member=job.spec.effective_min_available,
                    min_task_member={t.name: t.min_needed
                                     for t in job.spec.tasks},
                    queue=job.spec.queue,
                    priority_class=job.spec.priority_class,
                    min_resources=self._min_resources(job),
                    network_topology=job.spec.network_topology))
            self.store.create("PodGroup", pg)

    def _make_pod(self, job: Job, ts, index: int) -> Pod:
        tmpl = ts.template or {}
        pod = Pod(
            meta=ObjectMeta(
                name=self._pod_name(job, ts.name, index),
                namespace=job.meta.namespace,
                labels={LBL_JOB_NAME: job.meta.name, LBL_TASK_SPEC: ts.name,
                        LBL_TASK_INDEX: str(index)},
                annotations={ANN_PODGROUP: job.meta.name,
                             ANN_QUEUE: job.spec.queue},
                owner=f"Job/{job.meta.key}"),
            request=Resource.from_spec(tmpl.get("resources", {})),
            priority=int(tmpl.get("priority", 0)),
            priority_class=job.spec.priority_class,
            node_selector=dict(tmpl.get("node_selector", {})),
            tolerations=[Toleration(**t) for t in tmpl.get("tolerations", [])],
            affinity=tmpl.get("affinity"))
        for name, args in (job.spec.plugins or {}).items():
            fn = JOB_PLUGINS.get(name)
            if fn is not None:
                fn(job, ts, pod, index, args)
        return pod

    def sync_job(self, job: Job) -> None:
        phase = job.status.phase
        if phase in FINAL_PHASES:
            return
        if phase in (JobPhase.ABORTING.value, JobPhase.TERMINATING.value,
                     JobPhase.COMPLETING.value, JobPhase.RESTARTING.value):
            self._drain(job)
            return

        self._ensure_podgroup(job)
        pods = {p.meta.name: p for p in self._job_pods(job)}

        def dep_satisfied(ts) -> bool:
            """Task-level dependsOn (reference job.go TaskSpec.DependsOn):
            a task's pods are created only after each dependency task has
            its minimum running."""
            for dep in ts.depends_on:
                dep_ts = next((t for t in job.spec.tasks if t.name == dep),
                              None)
                if dep_ts is None:
                    continue
                running = sum(
                    1 for p in pods.values()
                    if p.meta.labels.get(LBL_TASK_SPEC) == dep
                    and p.phase == "Running")
                if running < dep_ts.min_needed:
                    return False
            return True

        # create missing / delete excess pods per task
        for ts in job.spec.tasks:
            if ts.depends_on and not dep_satisfied(ts):
                continue
            for i in range(ts.replicas):
                name = self._pod_name(job, ts.name, i)
                if name not in pods:
                    self.store.create("Pod", self._make_pod(job, ts, i))
            i = ts.replicas
            while True:
                name = self._pod_name(job, ts.name, i)
                if name in pods:
                    self.store.delete("Pod", job.meta.namespace, name)
                    i += 1
                else:
                    break

        self._update_status(job)

    def _update_status(self, job: Job) -> None:
        pods = self._job_pods(job)
        st = job.status
        old = (st.pending, st.running, st.succeeded, st.failed, st.phase)
        st.pending = sum(1 for p in pods if p.phase == "Pending")
        st.running = sum(1 for p in pods if p.phase == "Running")
        st.succeeded = sum(1 for p in pods if p.phase == "Succeeded")
        st.failed = sum(1 for p in pods if p.phase == "Failed")
        total = job.spec.total_replicas
        min_avail = job.spec.effective_min_available
        min_success = job.spec.min_success or total

        phase = st.phase
        if st.succeeded >= min_success:
            phase = JobPhase.COMPLETED.value
            self._fire_completed(job)
        elif st.failed > 0 and total - st.failed < min_avail:
            # can never reach the gang minimum again
            phase = JobPhase.FAILED.value
        elif st.running >= min_avail:
            phase = JobPhase.RUNNING.value
        st.phase = phase
        if (st.pending, st.running, st.succeeded, st.failed, st.phase) != old:
            self.store.update("Job", job)

    def _fire_completed(self, job: Job) -> None:
        pg = self.store.get("PodGroup", job.meta.namespace, job.meta.name)
        if pg is not None and pg.status.phase != PodGroupPhase.COMPLETED.value:
            pg.status.phase = PodGroupPhase.COMPLETED.value
            self.store.update("PodGroup", pg)

    def _drain(self, job: Job) -> None:
        """Aborting/Terminating/Completing/Restarting: delete pods, then
        settle into the target phase."""
        pods = self._job_pods(job)
        for p in pods:
            self.store.delete("Pod", p.meta.namespace, p.meta.name)
        nxt = {
            JobPhase.ABORTING.value: JobPhase.ABORTED.value,
            JobPhase.TERMINATING.value: JobPhase.TERMINATED.value,
            JobPhase.COMPLETING.value: JobPhase.COMPLETED.value,
            JobPhase.RESTARTING.value: JobPhase.PENDING.value,
        }[job.status.phase]
        job.status.phase = nxt
        if nxt == JobPhase.PENDING.value:
            self._dirty.add(job.meta.key)      # recreate pods next pass
        self.store.update("Job", job)

    def _cleanup_job(self, job: Job) -> None:
        for p in self._job_pods(job):
            self.store.delete("Pod", p.meta.namespace, p.meta.name)
        self.store.delete("PodGroup", job.meta.namespace, job.meta.name)
