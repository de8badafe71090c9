"""Controller suite: job state machine, lifecycle policies, podgroup
auto-creation, queue status, cronjob, jobflow DAG, garbage collection.

Mirrors the reference's controller unit tests (pkg/controllers/*/
*_test.go) driven through the in-process store.
"""

import time

import pytest

from volcano_amd.api.objects import (Command, CronJob, Job, JobFlow, FlowStep,
                                     JobSpec, JobTemplate, LifecyclePolicy,
                                     ObjectMeta, TaskSpec)
from volcano_amd.api.types import JobPhase, PodGroupPhase
from volcano_amd.controllers import ControllerManager
from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.kubelet import FakeKubelet

GI = 1024 ** 3


def mk_world(n_nodes=4):
    store = ObjectStore()
    for n in synth.make_nodes(n_nodes, cpu_milli=8000, mem=32 * GI):
        store.create("Node", n)
    cm = ControllerManager(store, ["job", "podgroup", "queue",
                                   "garbagecollector", "cronjob", "jobflow"])
    cache = SchedulerCache(store=store)
    sched = Scheduler(cache)
    kubelet = FakeKubelet(store)
    return store, cm, sched, kubelet


def mk_job(name, replicas=2, cpu="1", min_available=None, policies=None,
           plugins=None, tasks=None, **spec_kw):
    task_list = tasks or [TaskSpec(name="worker", replicas=replicas,
                                   template={"resources": {"cpu": cpu,
                                                           "memory": "1Gi"}})]
    return Job(meta=ObjectMeta(name=name),
               spec=JobSpec(tasks=task_list, min_available=min_available,
                            policies=policies or [], plugins=plugins or {},
                            **spec_kw))


def test_job_to_running_lifecycle():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job("j1", replicas=3))
    cm.sync_until_quiet()
    # controller created podgroup + pods
    assert store.get("PodGroup", "default", "j1") is not None
    assert store.count("Pod") == 3
    # scheduler binds the gang
    sched.run_once()
    assert all(p.node_name for p in store.list("Pod"))
    # kubelet runs them; job goes Running
    kubelet.tick()
    cm.sync_until_quiet()
    job = store.get("Job", "default", "j1")
    assert job.status.running == 3
    assert job.status.phase == JobPhase.RUNNING.value


def test_job_completion():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job("j2", replicas=2))
    cm.sync_until_quiet()
    sched.run_once()
    kubelet.tick()
    kubelet.tick(complete=lambda p: "Succeeded")
    cm.sync_until_quiet()
    job = store.get("Job", "default", "j2")
    assert job.status.phase == JobPhase.COMPLETED.value
    pg = store.get("PodGroup", "default", "j2")
    assert pg.status.phase == PodGroupPhase.COMPLETED.value


def test_job_ttl_garbage_collected():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job("j2b", replicas=2,
                               ttl_seconds_after_finished=0.0))
    cm.sync_until_quiet()
    sched.run_once()
    kubelet.tick()
    kubelet.tick(complete=lambda p: "Succeeded")
    cm.sync_until_quiet()
    # ttl=0 → GC removed the finished job (cascade: pods, podgroup)
    assert store.get("Job", "default", "j2b") is None
    assert store.count("Pod") == 0
    assert store.get("PodGroup", "default", "j2b") is None


def test_pod_failed_restart_policy_max_retry():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job(
        "j3", replicas=2, max_retry=1,
        policies=[LifecyclePolicy(events=["PodFailed"],
                                  action="RestartJob")]))
    cm.sync_until_quiet()
    sched.run_once()
    kubelet.tick()
    # one pod fails → RestartJob (retry 1)
    pod = store.list("Pod")[0]
    pod.phase = "Failed"
    store.update("Pod", pod)
    cm.sync_until_quiet()
    job = store.get("Job", "default", "j3")
    assert job.status.retry_count == 1
    assert job.status.phase == JobPhase.PENDING.value
    # pods were recreated fresh
    assert store.count("Pod") == 2
    assert all(p.phase == "Pending" for p in store.list("Pod"))
    # fail again → maxRetry exceeded → Failed
    sched2 = Scheduler(SchedulerCache(store=store))
    sched2.run_once()
    kubelet.tick()
    pod = store.list("Pod")[0]
    pod.phase = "Failed"
    store.update("Pod", pod)
    cm.sync_until_quiet()
    job = store.get("Job", "default", "j3")
    assert job.status.phase == JobPhase.FAILED.value


def test_abort_and_resume_via_command():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job("j4", replicas=2))
    cm.sync_until_quiet()
    store.create("Command", Command(meta=ObjectMeta(name="cmd1"),
                                    action="AbortJob", target_kind="Job",
                                    target_name="j4"))
    cm.sync_until_quiet()
    job = store.get("Job", "default", "j4")
    assert job.status.phase == JobPhase.ABORTED.value
    assert store.count("Pod") == 0
    store.create("Command", Command(meta=ObjectMeta(name="cmd2"),
                                    action="ResumeJob", target_kind="Job",
                                    target_name="j4"))
    cm.sync_until_quiet()
    job = store.get("Job", "default", "j4")
    assert job.status.phase == JobPhase.PENDING.value
    assert store.count("Pod") == 2


def test_job_plugins_inject_contracts():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job(
        "dist", plugins={"env": [], "svc": [], "pytorch": ["master"]},
        tasks=[TaskSpec(name="master", replicas=1,
                        template={"resources": {"cpu": "1"}}),
               TaskSpec(name="worker", replicas=2,
                        template={"resources": {"cpu": "1"}})]))
    cm.sync_until_quiet()
    pods = {p.meta.name: p for p in store.list("Pod")}
    w1 = pods["dist-worker-1"]
    assert w1.meta.annotations["env/VC_TASK_INDEX"] == "1"
    assert w1.meta.annotations["env/MASTER_ADDR"] == "dist-master-0.dist"
    assert w1.meta.annotations["env/RANK"] == "2"
    assert w1.meta.annotations["env/WORLD_SIZE"] == "3"
    assert "dist-master-0.dist" in w1.meta.annotations["svc/hosts"]


def test_podgroup_controller_wraps_normal_pod():
    store, cm, sched, kubelet = mk_world()
    store.create("Pod", synth.make_pod("lone", podgroup="", cpu_milli=500,
                                       mem=GI))
    pod = store.get("Pod", "default", "lone")
    pod.meta.annotations.pop("scheduling.volcano.sh/group-name", None)
    store.update("Pod", pod)
    cm.sync_until_quiet()
    pod = store.get("Pod", "default", "lone")
    pg_name = pod.meta.annotations.get("scheduling.volcano.sh/group-name")
    assert pg_name
    pg = store.get("PodGroup", "default", pg_name)
    assert pg is not None and pg.spec.min_member == 1
    # and it schedules
    sched.run_once()
    assert store.get("Pod", "default", "lone").node_name


def test_queue_status_counts():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job("q1", replicas=2))
    cm.sync_until_quiet()
    sched.run_once()
    kubelet.tick()
    cm.sync_until_quiet()
    q = store.get("Queue", "default", "default")
    assert q.status.running == 1


def test_cronjob_schedules_and_forbid():
    store, cm, sched, kubelet = mk_world()
    cron = cm.controllers[[c.name for c in cm.controllers].index("cronjob")]
    now = time.time()
    store.create("CronJob", CronJob(
        meta=ObjectMeta(name="cj"), schedule="* * * * *",
        concurrency_policy="Forbid",
        job_template=JobSpec(tasks=[TaskSpec(name="w", replicas=1,
                                             template={"resources": {"cpu": "1"}})])))
    cj = store.get("CronJob", "default", "cj")
    cj.meta.creation_timestamp = now - 120
    store.update("CronJob", cj)
    cron.now = now
    cm.sync_until_quiet()
    jobs = store.list("Job")
    assert len(jobs) == 1
    # next minute, previous job still active + Forbid → no new job
    cron.now = now + 61
    cm.sync_until_quiet()
    assert len(store.list("Job")) == 1


def test_jobflow_dag():
    store, cm, sched, kubelet = mk_world()
    spec = JobSpec(tasks=[TaskSpec(name="w", replicas=1,
                                   template={"resources": {"cpu": "1"}})])
    store.create("JobTemplate", JobTemplate(meta=ObjectMeta(name="a"), spec=spec))
    store.create("JobTemplate", JobTemplate(meta=ObjectMeta(name="b"), spec=spec))
    store.create("JobFlow", JobFlow(
        meta=ObjectMeta(name="flow"),
        flows=[FlowStep(name="a"),
               FlowStep(name="b", depends_on=["a"])]))
    cm.sync_until_quiet()
    assert store.get("Job", "default", "flow-a") is not None
    assert store.get("Job", "default", "flow-b") is None   # waits on a
    # run a to completion
    sched.run_once()
    kubelet.tick()
    kubelet.tick(complete=lambda p: "Succeeded")
    cm.sync_until_quiet()
    assert store.get("Job", "default", "flow-a").status.phase == \
        JobPhase.COMPLETED.value
    assert store.get("Job", "default", "flow-b") is not None
    flow = store.get("JobFlow", "default", "flow")
    assert flow.status["state"] == "Running"


def test_jobflow_cycle_detected():
    store, cm, sched, kubelet = mk_world()
    store.create("JobFlow", JobFlow(
        meta=ObjectMeta(name="bad"),
        flows=[FlowStep(name="a", depends_on=["b"]),
               FlowStep(name="b", depends_on=["a"])]))
    cm.sync_until_quiet()
    assert store.get("JobFlow", "default", "bad").status["state"] == "Failed"


def test_task_depends_on():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job(
        "dep", min_available=1,
        tasks=[TaskSpec(name="ps", replicas=1,
                        template={"resources": {"cpu": "1"}}),
               TaskSpec(name="worker", replicas=2, depends_on=["ps"],
                        template={"resources": {"cpu": "1"}})]))
    cm.sync_until_quiet()
    # only ps pods exist until ps is Running
    assert {p.meta.name for p in store.list("Pod")} == {"dep-ps-0"}
    sched.run_once()
    kubelet.tick()        # ps starts Running
    cm.sync_until_quiet()
    names = {p.meta.name for p in store.list("Pod")}
    assert names == {"dep-ps-0", "dep-worker-0", "dep-worker-1"}
    sched.run_once()
    assert all(p.node_name for p in store.list("Pod"))


def test_hyperjob_expands_and_aggregates():
    store, cm, sched, kubelet = mk_world()
    from volcano_amd.api.objects import HyperJob, JobSpec, TaskSpec
    from volcano_amd.controllers import ControllerManager
    cm2 = ControllerManager(store, ["job", "podgroup", "queue", "hyperjob"])
    store.create("HyperJob", HyperJob(
        meta=ObjectMeta(name="hj"), replicas=2,
        job_template=JobSpec(tasks=[TaskSpec(
            name="w", replicas=1,
            template={"resources": {"cpu": "1"}})])))
    cm2.sync_until_quiet()
    assert store.get("Job", "default", "hj-0") is not None
    assert store.get("Job", "default", "hj-1") is not None
    sched.run_once()
    kubelet.tick()
    kubelet.tick(complete=lambda p: "Succeeded")
    cm2.sync_until_quiet()
    hj = store.get("HyperJob", "default", "hj")
    assert hj.status["state"] == "Completed"


def test_elastic_scale_up_down():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job("el", replicas=2, min_available=1))
    cm.sync_until_quiet()
    sched.run_once()
    kubelet.tick()
    cm.sync_until_quiet()
    assert store.count("Pod") == 2
    # scale up
    job = store.get("Job", "default", "el")
    job.spec.tasks[0].replicas = 5
    store.update("Job", job)
    cm.sync_until_quiet()
    assert store.count("Pod") == 5
    sched.run_once()
    assert sum(1 for p in store.list("Pod") if p.node_name) == 5
    # scale down
    job = store.get("Job", "default", "el")
    job.spec.tasks[0].replicas = 3
    store.update("Job", job)
    cm.sync_until_quiet()
    assert store.count("Pod") == 3


def test_queue_close_drains_then_closes():
    store, cm, sched, kubelet = mk_world()
    store.create("Job", mk_job("qc", replicas=1,
                               ttl_seconds_after_finished=0.0))
    cm.sync_until_quiet()
    store.create("Command", Command(meta=ObjectMeta(name="close-default"),
                                    action="CloseQueue", target_kind="Queue",
                                    target_name="default"))
    cm.sync_until_quiet()
    q = store.get("Queue", "default", "default")
    assert q.status.state == "Closing"       # podgroup still exists
    # closed queues admit nothing new
    sched.run_once()
    assert all(not p.node_name for p in store.list("Pod"))
    # drain: job finishes and is GC'd → queue transitions Closed
    job = store.get("Job", "default", "qc")
    job.status.phase = "Completed"
    store.update("Job", job)
    cm.sync_until_quiet()
    q = store.get("Queue", "default", "default")
    assert q.status.state == "Closed"
    # reopen works
    store.create("Command", Command(meta=ObjectMeta(name="open-default"),
                                    action="OpenQueue", target_kind="Queue",
                                    target_name="default"))
    cm.sync_until_quiet()
    assert store.get("Queue", "default",
                     "default").status.state == "Open"


def test_job_controller_sharded_workers_and_error_backoff():
    """Reference job_controller.go: hashed worker shards + errTasks
    rate-limited resync."""
    import time

    from volcano_amd.controllers.job import JobController
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    store = ObjectStore()
    store.create("Queue", synth.make_queue("default"))
    ctl = JobController()
    ctl.initialize(store)
    # distinct jobs land on distinct shards deterministically
    shards = {JobController._shard_of(f"default/j{i}", 4)
              for i in range(32)}
    assert len(shards) > 1
    for i in range(8):
        store.create("Job", mk_job(f"shard-{i}", replicas=2))
    ctl.sync_once()
    # all jobs reconciled (pods created) despite multi-shard execution
    for i in range(8):
        pods = [p for p in store.list("Pod")
                if p.meta.labels.get("volcano.sh/job-name") == f"shard-{i}"]
        assert len(pods) == 2

    # error backoff: a failing sync re-queues with increasing delay
    fails = {"n": 0}
    orig = ctl.sync_job

    def flaky(job):
        if job.meta.name == "shard-0" and fails["n"] < 2:
            fails["n"] += 1
            raise RuntimeError("transient")
        return orig(job)

    ctl.sync_job = flaky
    ctl._dirty.add("default/shard-0")
    ctl.sync_once()
    assert fails["n"] == 1
    assert "default/shard-0" in ctl._err_queue
    ctl.sync_once()                      # backoff not yet expired
    assert fails["n"] == 1
    time.sleep(0.06)
    ctl.sync_once()                      # first retry (fails again)
    assert fails["n"] == 2
    time.sleep(0.11)
    ctl.sync_once()                      # second retry succeeds
    assert "default/shard-0" not in ctl._err_queue


def test_jobtemplate_controller_bookkeeping():
    """Standalone jobtemplate controller (reference
    pkg/controllers/jobtemplate): status tracks jobs created from the
    template."""
    from volcano_amd.api.objects import (FlowStep, JobFlow, JobTemplate,
                                         ObjectMeta)
    from volcano_amd.controllers import ControllerManager
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    store = ObjectStore()
    store.create("Queue", synth.make_queue("default"))
    store.create("JobTemplate", JobTemplate(
        meta=ObjectMeta(name="step-a", namespace="default"),
        spec=mk_job("ignored", replicas=1).spec))
    store.create("JobFlow", JobFlow(
        meta=ObjectMeta(name="f1", namespace="default"),
        flows=[FlowStep(name="step-a")]))
    cm = ControllerManager(store, ["jobflow", "jobtemplate", "job",
                                   "podgroup"])
    cm.sync_until_quiet()
    tmpl = store.get("JobTemplate", "default", "step-a")
    assert tmpl.status.get("jobDependsOnList") == ["f1-step-a"]


def test_datasource_claim_binding_and_reclaim():
    """datadependency/v1alpha1 (reference staging types.go:32-201):
    claims bind by (system, type, name); the source mirrors claimRefs;
    reclaimPolicy=Delete removes the source when the last claim goes."""
    from volcano_amd.api.objects import DataSource, DataSourceClaim, ObjectMeta
    from volcano_amd.controllers import ControllerManager
    from volcano_amd.store import ObjectStore

    store = ObjectStore()
    mgr = ControllerManager(store, controllers=["datadependency"])
    src = DataSource(meta=ObjectMeta(name="sales", namespace=""),
                     system="hive", type="table", name="db.sales",
                     cluster_names=["c1"], reclaim_policy="Delete")
    store.create("DataSource", src)
    claim = DataSourceClaim(
        meta=ObjectMeta(name="train-input", namespace="ml"),
        system="hive", data_source_type="table",
        data_source_name="db.sales",
        workload={"apiVersion": "batch.volcano.sh/v1alpha1",
                  "kind": "Job", "name": "train"})
    store.create("DataSourceClaim", claim)
    mgr.sync_once()

    c = store.get("DataSourceClaim", "ml", "train-input")
    assert c.phase == "Bound" and c.bound_data_source == "sales"
    s = store.get("DataSource", "", "sales")
    assert s.bound_claims == 1 and s.claim_refs == ["ml/train-input"]

    # unmatched claim stays Pending
    orphan = DataSourceClaim(
        meta=ObjectMeta(name="orphan", namespace="ml"),
        system="s3", data_source_type="bucket", data_source_name="nope")
    store.create("DataSourceClaim", orphan)
    mgr.sync_once()
    assert store.get("DataSourceClaim", "ml", "orphan").phase == "Pending"

    # last claim released → Delete reclaim removes the source
    store.delete("DataSourceClaim", "ml", "train-input")
    mgr.sync_once()
    assert store.get("DataSource", "", "sales") is None


def test_job_state_action_matrix():
    """Reference job_state_test.go per-state action sets: finished jobs
    ignore every action; Aborted acts only on Resume (→ Restarting with
    a retry bump); live phases accept the full action set."""
    from volcano_amd.api.types import Action
    store, cm, sched, kubelet = mk_world()
    jc = next(c for c in cm.controllers if c.__class__.__name__
              == "JobController")

    def job_in(phase):
        j = mk_job("sm", replicas=1)
        j.status.phase = phase
        store.create("Job", j)
        return j

    # finished states: every action is a no-op
    for phase in ("Completed", "Failed", "Terminated"):
        j = job_in(phase)
        for act in ("AbortJob", "RestartJob", "CompleteJob",
                    "TerminateJob", "ResumeJob"):
            jc.execute_action(j, act)
            assert j.status.phase == phase, (phase, act)
        store.delete("Job", "default", "sm")

    # Aborted: only Resume acts — Restarting with retry bump
    j = job_in("Aborted")
    for act in ("AbortJob", "RestartJob", "CompleteJob", "TerminateJob"):
        jc.execute_action(j, act)
        assert j.status.phase == "Aborted", act
    jc.execute_action(j, "ResumeJob")
    assert j.status.phase == "Restarting"
    assert j.status.retry_count == 1
    store.delete("Job", "default", "sm")

    # live phases take the whole action set
    for act, want in (("AbortJob", "Aborting"),
                      ("TerminateJob", "Terminating"),
                      ("CompleteJob", "Completing")):
        j = job_in("Running")
        jc.execute_action(j, act)
        assert j.status.phase == want, act
        store.delete("Job", "default", "sm")
