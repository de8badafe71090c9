"""Scheduler-in-a-box: store → cache → session → actions → binds.

Port of the reference's uthelper pattern (pkg/scheduler/uthelper/
helper.go): declare nodes/podgroups/pods/queues, run the real actions,
assert the bind map.
"""

import pytest

from volcano_amd.api.resource import CPU, MEMORY
from volcano_amd.api.types import PodGroupPhase, TaskStatus
from volcano_amd.api.objects import Taint, Toleration
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk(store=None, binder=None):
    store = store or ObjectStore()
    binder = binder or FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    sched = Scheduler(cache)
    return store, binder, cache, sched


def test_single_gang_job_schedules():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(4, cpu_milli=4000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "j1", replicas=3, cpu_milli=1000, mem=1 * GI)

    ssn = sched.run_once()
    assert len(binder.binds) == 3
    job = cache.jobs["default/j1"]
    assert job.is_ready()
    assert job.podgroup.status.phase == PodGroupPhase.RUNNING.value
    # all bound tasks landed on real nodes with capacity accounted
    for tkey, node in binder.binds.items():
        assert node.startswith("node-")
    total_used = sum(ni.used.milli_cpu for ni in cache.nodes.values())
    assert total_used == 3000


def test_gang_insufficient_capacity_reverts():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(2, cpu_milli=2000, mem=4 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    # needs 5×1cpu but cluster fits only 4 → gang must NOT partially place
    synth.make_gang(store, "big", replicas=5, cpu_milli=1000, mem=1 * GI)

    sched.run_once()
    assert binder.binds == {}
    job = cache.jobs["default/big"]
    assert job.occupied_count == 0
    for ni in cache.nodes.values():
        assert ni.used.milli_cpu == 0


def test_partial_gang_min_member():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(2, cpu_milli=2000, mem=4 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    # 5 replicas, minMember 3 → exactly 4 fit, gang satisfied
    synth.make_gang(store, "elastic", replicas=5, min_member=3,
                    cpu_milli=1000, mem=1 * GI)
    sched.run_once()
    assert len(binder.binds) == 4


def test_two_jobs_fifo_priority():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(1, cpu_milli=3000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "low", replicas=3, cpu_milli=1000, mem=1 * GI,
                    priority=1)
    synth.make_gang(store, "high", replicas=3, cpu_milli=1000, mem=1 * GI,
                    priority=10)
    sched.run_once()
    # only one gang fits; priority plugin must pick "high"
    assert len(binder.binds) == 3
    assert all(k.startswith("default/high") for k in binder.binds)


def test_node_selector_and_taints():
    store, binder, cache, sched = mk()
    store.create("Node", synth.make_node("gpu-node", cpu_milli=8000,
                                         mem=16 * GI,
                                         labels={"accel": "mi355x"}))
    store.create("Node", synth.make_node("cpu-node", cpu_milli=8000,
                                         mem=16 * GI))
    store.create("Node", synth.make_node("tainted", cpu_milli=8000,
                                         mem=16 * GI,
                                         taints=[Taint("dedicated", "infra",
                                                       "NoSchedule")]))
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "sel", replicas=2, cpu_milli=500, mem=GI,
                    node_selector={"accel": "mi355x"})
    sched.run_once()
    assert len(binder.binds) == 2
    assert set(binder.binds.values()) == {"gpu-node"}

    # intolerant pods never land on the tainted node
    synth.make_gang(store, "any", replicas=6, cpu_milli=2000, mem=GI)
    sched.run_once()
    landed = {binder.binds[k] for k in binder.binds if k.startswith("default/any")}
    assert "tainted" not in landed

    # tolerant pods may
    synth.make_gang(store, "tol", replicas=1, cpu_milli=4000, mem=GI,
                    tolerations=[Toleration(key="dedicated", value="infra",
                                            effect="NoSchedule")])
    sched.run_once()
    assert "default/tol-worker-0" in binder.binds


def test_multi_role_min_task_member():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(3, cpu_milli=4000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("mr", min_member=3,
                             min_task_member={"ps": 1, "worker": 2})
    store.create("PodGroup", pg)
    store.create("Pod", synth.make_pod("mr-ps-0", "mr", role="ps",
                                       cpu_milli=2000, mem=GI))
    for i in range(2):
        store.create("Pod", synth.make_pod(f"mr-worker-{i}", "mr",
                                           role="worker", cpu_milli=1000,
                                           mem=GI))
    sched.run_once()
    assert len(binder.binds) == 3


def test_multi_role_gang_fails_atomically():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(1, cpu_milli=3000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("mr2", min_member=3,
                             min_task_member={"ps": 1, "worker": 2})
    store.create("PodGroup", pg)
    # ps needs 4 cpu — can't fit; whole gang must revert (workers too)
    store.create("Pod", synth.make_pod("mr2-ps-0", "mr2", role="ps",
                                       cpu_milli=4000, mem=GI))
    for i in range(2):
        store.create("Pod", synth.make_pod(f"mr2-worker-{i}", "mr2",
                                           role="worker", cpu_milli=1000,
                                           mem=GI))
    sched.run_once()
    assert binder.binds == {}
    for ni in cache.nodes.values():
        assert ni.used.milli_cpu == 0


def test_queue_proportion_limits():
    store, binder, cache, sched = mk()
    # 10 cpus total; queues weighted 1:1 → 5 cpu deserved each
    for n in synth.make_nodes(1, cpu_milli=10000, mem=64 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("qa", weight=1))
    store.create("Queue", synth.make_queue("qb", weight=1))
    synth.make_gang(store, "ja", replicas=8, min_member=1, queue="qa",
                    cpu_milli=1000, mem=GI)
    synth.make_gang(store, "jb", replicas=8, min_member=1, queue="qb",
                    cpu_milli=1000, mem=GI)
    sched.run_once()
    ja = sum(1 for k in binder.binds if k.startswith("default/ja"))
    jb = sum(1 for k in binder.binds if k.startswith("default/jb"))
    # each queue capped at its 5-cpu deserved share
    assert ja == 5 and jb == 5


def test_best_effort_backfill():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(1, cpu_milli=1000, mem=GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "be", replicas=2, cpu_milli=0, mem=0)
    sched.run_once()
    assert len(binder.binds) == 2


def test_bound_pods_survive_restart():
    """Crash-resume: rebuild cache from store only (reference: informers
    are the only durable state, SURVEY §5 checkpoint/resume)."""
    from volcano_amd.scheduler.cache import StoreBinder
    store = ObjectStore()
    store_binder = None
    for n in synth.make_nodes(2, cpu_milli=2000, mem=4 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "j1", replicas=2, cpu_milli=1000, mem=GI)
    cache = SchedulerCache(store=store)   # default StoreBinder: persists binds
    sched = Scheduler(cache)
    sched.run_once()
    bound = [p for p in store.list("Pod") if p.node_name]
    assert len(bound) == 2

    # new scheduler process over the same store
    binder2 = FakeBinder()
    cache2 = SchedulerCache(store=store, binder=binder2)
    sched2 = Scheduler(cache2)
    sched2.run_once()
    # nothing new to bind; node usage reconstructed
    assert binder2.binds == {}
    used = sum(ni.used.milli_cpu for ni in cache2.nodes.values())
    assert used == 2000


def test_fifo_dequeue_blocks_queue():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(1, cpu_milli=10000, mem=64 * GI):
        store.create("Node", n)
    q = synth.make_queue("fq")
    q.spec.dequeue_strategy = "fifo"
    q.spec.capability = __import__("volcano_amd.api.resource",
                                   fromlist=["Resource"]).Resource(
        {"cpu": 4000.0})
    store.create("Queue", q)
    # head job too big even for overcommitted capacity → enqueue rejects
    # it, and fifo blocks the line behind it
    synth.make_gang(store, "big-head", replicas=16, queue="fq",
                    cpu_milli=1000, mem=GI)
    import time as _t
    _t.sleep(0.01)
    synth.make_gang(store, "small", replicas=1, queue="fq", cpu_milli=1000,
                    mem=GI)
    sched.run_once()
    assert binder.binds == {}       # fifo: small waits behind big-head

    # traverse queue admits the small job past the stuck head
    q.spec.dequeue_strategy = "traverse"
    store.update("Queue", q)
    sched.run_once()
    assert any(k.startswith("default/small") for k in binder.binds)


def test_bundle_slot_recycling_quota_starved_head():
    """Regression: a quota-starved big gang bundled with a small identical
    job must not drag the small job down — its slots recycle."""
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(1, cpu_milli=10000, mem=64 * GI):
        store.create("Node", n)
    q = synth.make_queue("rq")
    from volcano_amd.api.resource import Resource
    q.spec.capability = Resource({"cpu": 4000.0})
    store.create("Queue", q)
    # head: 8-gang (min 8) > queue quota 4 → cannot place
    synth.make_gang(store, "head", replicas=8, queue="rq", cpu_milli=1000,
                    mem=GI)
    import time
    time.sleep(0.01)
    # small identical-signature 1-gang bundles with it
    synth.make_gang(store, "tiny", replicas=1, queue="rq", cpu_milli=1000,
                    mem=GI)
    sched.run_once()
    assert "default/tiny-worker-0" in binder.binds
    assert not any(k.startswith("default/head") for k in binder.binds)
    # staged usage fully unwound for the failed gang
    assert sum(ni.used.milli_cpu for ni in cache.nodes.values()) == 1000


def test_multi_value_node_affinity_in():
    store, binder, cache, sched = mk()
    for name, zone in [("a", "z1"), ("b", "z2"), ("c", "z3")]:
        store.create("Node", synth.make_node(name, cpu_milli=4000,
                                             mem=8 * GI,
                                             labels={"zone": zone}))
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("mv", min_member=2)
    store.create("PodGroup", pg)
    for i in range(2):
        pod = synth.make_pod(f"mv-w-{i}", "mv", cpu_milli=1000, mem=GI)
        pod.affinity = {"in": {"zone": ["z1", "z3"]}}   # OR over values
        store.create("Pod", pod)
    sched.run_once()
    assert len(binder.binds) == 2
    assert set(binder.binds.values()) <= {"a", "c"}


def test_engine_idle_fast_path():
    store, binder, cache, sched = mk()
    for n in synth.make_nodes(1, cpu_milli=4000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "ij", replicas=2, cpu_milli=1000, mem=GI)
    assert not sched._idle()          # pending work + fresh events
    sched.run_once()
    assert len(binder.binds) == 2
    sched._idle()                     # drains the cycle's own store writes
    assert sched._idle()              # everything bound, no new events
    synth.make_gang(store, "ij2", replicas=1, cpu_milli=1000, mem=GI)
    assert not sched._idle()          # new events wake it
