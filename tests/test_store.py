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
