"""Object store: CRUD, watch replay, durability round-trip."""

import os

from volcano_amd.api.objects import ObjectMeta, Queue
from volcano_amd.store import EventType, ObjectStore
from volcano_amd.utils.synth import make_node, make_pod, make_podgroup


def test_crud_and_versions():
    s = ObjectStore()
    n = make_node("n1")
    s.create("Node", n)
    assert s.get("Node", "default", "n1") is n
    rv1 = n.meta.resource_version
    s.update("Node", n)
    assert n.meta.resource_version > rv1
    assert s.count("Node") == 1
    s.delete("Node", "default", "n1")
    assert s.get("Node", "default", "n1") is None


def test_watch_replay_and_stream():
    s = ObjectStore()
    s.create("Node", make_node("n1"))
    w = s.watch("Node", "Pod")
    evs = w.drain()
    assert len(evs) == 1 and evs[0].type == EventType.ADDED
    s.create("Pod", make_pod("p1", "pg1"))
    s.create("Queue", Queue(meta=ObjectMeta(name="q")))  # not watched
    evs = w.drain()
    assert len(evs) == 1 and evs[0].kind == "Pod"
    w.stop()


def test_save_load_roundtrip(tmp_path):
    s = ObjectStore()
    s.create("Node", make_node("n1", cpu_milli=8000))
    s.create("PodGroup", make_podgroup("pg1", min_member=3))
    s.create("Pod", make_pod("p1", "pg1", cpu_milli=250))
    path = str(tmp_path / "state.json")
    s.save(path)
    s2 = ObjectStore.load(path)
    assert s2.count("Node") == 1
    pg = s2.get("PodGroup", "default", "pg1")
    assert pg.spec.min_member == 3
    pod = s2.get("Pod", "default", "p1")
    assert pod.request.milli_cpu == 250
    assert pod.podgroup_name == "pg1"


def test_watch_overflow_drops_and_flags():
    from volcano_amd.utils import synth
    store = ObjectStore()
    w = store.watch("Node", replay=False, maxsize=4)
    for i in range(10):
        store.create("Node", synth.make_node(f"ovf-{i}"))
    assert w.overflowed                    # writers never blocked
    assert len(w.drain()) == 4             # bounded queue kept the head
    w.stop()


def test_cache_resyncs_after_watch_overflow():
    from volcano_amd.scheduler import FakeBinder, SchedulerCache
    from volcano_amd.utils import synth
    store = ObjectStore()
    cache = SchedulerCache(store=store, binder=FakeBinder())
    cache._watch.stop()
    cache._watch = store.watch("Pod", "Node", "PodGroup", "Queue", maxsize=8)
    for i in range(30):
        store.create("Node", synth.make_node(f"rs-{i}"))
    synth.make_gang(store, "g", replicas=2, cpu_milli=100)
    assert cache._watch.overflowed
    cache.sync()
    # the relist rebuilt the complete world despite the dropped events
    assert len(cache.nodes) == 30
    assert "default/g" in cache.jobs
    assert not cache._watch.overflowed
