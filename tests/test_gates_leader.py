"""Scheduling gates + queue-admission gate lifting; leader election."""

from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.leaderelection import LeaderElector

GI = 1024 ** 3


def test_scheduling_gates_hold_then_lift():
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    sched = Scheduler(cache)
    for n in synth.make_nodes(2, cpu_milli=4000, mem=16 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("gj", min_member=2)
    store.create("PodGroup", pg)
    for i in range(2):
        pod = synth.make_pod(f"gj-w-{i}", "gj", cpu_milli=1000, mem=GI)
        pod.scheduling_gates = ["volcano.sh/queue-allocation-gate"]
        store.create("Pod", pod)
    sched.run_once()
    # first cycle: enqueue admits and lifts the queue gate; tasks become
    # schedulable within the same or next cycle
    bound = len(binder.binds)
    if bound == 0:
        sched.run_once()
        bound = len(binder.binds)
    assert bound == 2
    for p in store.list("Pod"):
        assert p.scheduling_gates == []


def test_custom_gate_keeps_holding():
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    sched = Scheduler(cache)
    for n in synth.make_nodes(1, cpu_milli=4000, mem=16 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("hold", min_member=1)
    store.create("PodGroup", pg)
    pod = synth.make_pod("hold-w-0", "hold", cpu_milli=1000, mem=GI)
    pod.scheduling_gates = ["example.com/custom-gate"]
    store.create("Pod", pod)
    sched.run_once()
    sched.run_once()
    assert binder.binds == {}       # only the queue gate is auto-lifted


def test_leader_election_exclusive(tmp_path):
    a = LeaderElector("test", lock_dir=str(tmp_path))
    b = LeaderElector("test", lock_dir=str(tmp_path))
    assert a.try_acquire()
    assert a.is_leader
    assert not b.try_acquire()
    a.release()
    assert b.try_acquire()
    b.release()
