"""Node-affinity operator coverage (Exists/DoesNotExist/Gt/Lt/multi-In)
and Statement merge/save/recover (reference framework/statement.go:433-483)."""

from volcano_amd.api.resource import CPU
from volcano_amd.api.types import TaskStatus
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.statement import Statement
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def world(node_labels):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    sched = Scheduler(cache, default_config())
    for name, labels in node_labels.items():
        store.create("Node", synth.make_node(
            name, cpu_milli=8000, mem=32 * GI, labels=labels))
    return store, binder, cache, sched


def run_one(store, binder, sched, affinity):
    synth.make_gang(store, "job", replicas=1, cpu_milli=500, mem=GI)
    pod = store.get("Pod", "default", "job-worker-0")
    pod.affinity = affinity
    store.update("Pod", pod)
    sched.run_once()
    return binder.binds.get("default/job-worker-0")


def test_affinity_exists():
    store, binder, cache, sched = world({
        "a": {"disk": "ssd"}, "b": {}})
    assert run_one(store, binder, sched, {"exists": ["disk"]}) == "a"


def test_affinity_not_exists():
    store, binder, cache, sched = world({
        "a": {"disk": "ssd"}, "b": {}})
    assert run_one(store, binder, sched, {"notExists": ["disk"]}) == "b"


def test_affinity_gt_lt():
    store, binder, cache, sched = world({
        "a": {"gen": "3"}, "b": {"gen": "7"}, "c": {}})
    assert run_one(store, binder, sched, {"gt": {"gen": 5}}) == "b"
    binder.binds.clear()
    store2, binder2, cache2, sched2 = world({
        "a": {"gen": "3"}, "b": {"gen": "7"}, "c": {}})
    assert run_one(store2, binder2, sched2, {"lt": {"gen": 5}}) == "a"


def test_affinity_multi_value_in():
    store, binder, cache, sched = world({
        "a": {"zone": "z1"}, "b": {"zone": "z2"}, "c": {"zone": "z3"}})
    got = run_one(store, binder, sched, {"in": {"zone": ["z2", "z3"]}})
    assert got in ("b", "c")


def test_affinity_infeasible_blocks():
    store, binder, cache, sched = world({"a": {}, "b": {}})
    assert run_one(store, binder, sched, {"exists": ["nvme"]}) is None


# -- Statement composition ---------------------------------------------------

def _session_with_bound_task():
    store, binder, cache, sched = world({"a": {}})
    synth.make_gang(store, "vic", replicas=1, cpu_milli=500, mem=GI)
    sched.run_once()
    ssn = sched.open_session()
    task = next(iter(ssn.jobs["default/vic"].tasks.values()))
    return sched, ssn, task


def test_statement_merge_moves_ops():
    sched, ssn, task = _session_with_bound_task()
    s1, s2 = Statement(ssn), Statement(ssn)
    s2.evict(task, "why")
    s1.merge(s2)
    assert s2.ops == [] and len(s1.ops) == 1 and s1.has_evictions()
    s1.discard()
    assert task.status != TaskStatus.RELEASING
    sched.close_session(ssn)


def test_statement_save_recover_roundtrip():
    sched, ssn, task = _session_with_bound_task()
    s = Statement(ssn)
    s.evict(task, "preempted")
    saved = s.save_operations()
    s.discard()                      # decisions undone this cycle…
    assert task.status != TaskStatus.RELEASING
    s2 = Statement(ssn)
    assert s2.recover_operations(saved)       # …replayed later
    assert task.status == TaskStatus.RELEASING
    s2.discard()
    sched.close_session(ssn)


def test_statement_recover_missing_task_rolls_back():
    sched, ssn, task = _session_with_bound_task()
    s = Statement(ssn)
    saved = [("evict", task.key, "x"), ("evict", "default/ghost", "x")]
    assert not s.recover_operations(saved)
    # first op must have been rolled back with the failure
    assert task.status != TaskStatus.RELEASING and s.ops == []
    sched.close_session(ssn)


# -- volume zone filter ------------------------------------------------------

def test_volume_zone_confines_pod():
    from volcano_amd.api.objects import (ObjectMeta, PersistentVolume,
                                         PersistentVolumeClaim, ZONE_LABEL)
    store, binder, cache, sched = world({
        "a": {ZONE_LABEL: "z1"}, "b": {ZONE_LABEL: "z2"}, "c": {}})
    store.create("PersistentVolume", PersistentVolume(
        meta=ObjectMeta(name="pv1", labels={ZONE_LABEL: "z2"}),
        capacity=10 * GI))
    store.create("PersistentVolumeClaim", PersistentVolumeClaim(
        meta=ObjectMeta(name="data", namespace="default"),
        volume_name="pv1"))
    synth.make_gang(store, "vj", replicas=1, cpu_milli=500, mem=GI)
    pod = store.get("Pod", "default", "vj-worker-0")
    pod.volumes = ["data"]
    store.update("Pod", pod)
    sched.run_once()
    assert binder.binds.get("default/vj-worker-0") == "b"   # PV's zone


def test_unbound_pvc_constrains_nothing():
    from volcano_amd.api.objects import (ObjectMeta, PersistentVolumeClaim)
    store, binder, cache, sched = world({"a": {}, "b": {}})
    store.create("PersistentVolumeClaim", PersistentVolumeClaim(
        meta=ObjectMeta(name="later", namespace="default")))  # unbound
    synth.make_gang(store, "uj", replicas=1, cpu_milli=500, mem=GI)
    pod = store.get("Pod", "default", "uj-worker-0")
    pod.volumes = ["later"]
    store.update("Pod", pod)
    sched.run_once()
    assert binder.binds.get("default/uj-worker-0") in ("a", "b")
