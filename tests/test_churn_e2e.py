"""Steady-state churn: jobs arrive, run, complete and are GC'd while the
scheduler cycles — the long-running-cluster shape (reference e2e
jobseq/stress analog, in-process)."""

import random

from volcano_amd.api.types import JobPhase
from volcano_amd.controllers import ControllerManager
from volcano_amd.scheduler import Scheduler, SchedulerCache
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.kubelet import FakeKubelet
from tests.test_controllers import mk_job

GI = 1024 ** 3


def test_churn_cycles():
    rng = random.Random(0)
    store = ObjectStore()
    for n in synth.make_nodes(6, cpu_milli=8000, mem=32 * GI):
        store.create("Node", n)
    cm = ControllerManager(store, ["job", "podgroup", "queue",
                                   "garbagecollector"])
    cache = SchedulerCache(store=store)
    sched = Scheduler(cache)
    kubelet = FakeKubelet(store)

    submitted = completed = 0
    ttl_done = set()
    for cycle in range(30):
        # new work arrives
        if cycle % 3 == 0:
            submitted += 1
            store.create("Job", mk_job(
                f"churn-{cycle:03d}", replicas=rng.randint(1, 4),
                cpu="1", ttl_seconds_after_finished=0.0))
        cm.sync_until_quiet()
        sched.run_once()
        # pods run; ~every other cycle some job finishes wholesale
        kubelet.tick()
        if cycle % 2 == 1:
            running_jobs = [j for j in store.list("Job")
                            if j.status.phase == JobPhase.RUNNING.value]
            if running_jobs:
                victim = rng.choice(running_jobs)
                name = victim.meta.name
                kubelet.tick(complete=lambda p, n=name:
                             "Succeeded" if p.meta.labels.get(
                                 "volcano.sh/job-name") == n else None)
        cm.sync_until_quiet()

    # nothing leaked: every completed job was GC'd (ttl=0); remaining jobs
    # are consistent with their pods
    jobs = store.list("Job")
    for j in jobs:
        assert j.status.phase in (JobPhase.PENDING.value,
                                  JobPhase.RUNNING.value)
    # node accounting consistent after 30 cycles of churn
    for ni in cache.nodes.values():
        recomputed = sum(t.request.milli_cpu for t in ni.tasks.values()
                         if t.status.occupies_node)
        assert abs(recomputed - ni.used.milli_cpu) < 1.0
        assert ni.used.milli_cpu <= ni.allocatable.milli_cpu + 0.5
    # and the scheduler kept making progress
    assert submitted >= 10
    bound_pods = [p for p in store.list("Pod") if p.node_name]
    assert bound_pods or submitted > len(jobs)   # some work flowed through
