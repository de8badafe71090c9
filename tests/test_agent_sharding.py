"""Node agent (oversubscription/eviction/QoS), agent-scheduler fast
path, sharding controller 2-phase handoff, colocation config, cache
dumper."""

import json

from volcano_amd.agent import (CpuQosHandler, EventsManager, EvictionHandler,
                               MemoryQosHandler, NetworkQosHandler,
                               OversubscriptionHandler)
from volcano_amd.agent.eventsmgr import NodeUsage
from volcano_amd.api.objects import ColocationConfig, ObjectMeta
from volcano_amd.api.resource import CPU
from volcano_amd.controllers import ControllerManager
from volcano_amd.controllers.sharding import ShardingController
from volcano_amd.scheduler import FakeBinder, Scheduler, SchedulerCache
from volcano_amd.scheduler.agentscheduler import AgentScheduler
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def test_oversubscription_annotation_and_scheduler_pickup():
    store = ObjectStore()
    node = synth.make_node("n1", cpu_milli=10000, mem=32 * GI)
    store.create("Node", node)
    mgr = EventsManager(store, "n1")
    mgr.register(OversubscriptionHandler(ratio=0.5))
    mgr.probe.injected = NodeUsage("n1", cpu_pct=20.0, mem_pct=20.0)
    mgr.tick()
    node = store.get("Node", "default", "n1")
    # oversold cpu = 10000 * 0.8 * 0.5 = 4000
    assert node.meta.annotations["volcano.sh/oversubscription-cpu"] == "4000"

    # scheduler sees 14 cpus of allocatable now
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    sched = Scheduler(cache)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "big", replicas=12, cpu_milli=1000, mem=GI)
    sched.run_once()
    assert len(binder.binds) == 12      # fits only with oversubscription

    # pressure: oversell retracts
    mgr.probe.injected = NodeUsage("n1", cpu_pct=95.0, mem_pct=50.0)
    mgr.tick()
    node = store.get("Node", "default", "n1")
    assert node.meta.annotations["volcano.sh/oversubscription-cpu"] == "0"


def test_eviction_handler_targets_offline_pods():
    store = ObjectStore()
    store.create("Node", synth.make_node("n1"))
    on = synth.make_pod("online", "g1", cpu_milli=2000, mem=GI,
                        node_name="n1", phase="Running")
    off = synth.make_pod("offline", "g2", cpu_milli=1000, mem=GI,
                         node_name="n1", phase="Running")
    off.meta.annotations["volcano.sh/preemptable"] = "true"
    store.create("Pod", on)
    store.create("Pod", off)
    mgr = EventsManager(store, "n1")
    mgr.register(EvictionHandler(high_watermark=90))
    mgr.probe.injected = NodeUsage("n1", cpu_pct=95.0, mem_pct=50.0)
    mgr.tick()
    assert store.get("Pod", "default", "offline").phase == "Failed"
    assert store.get("Pod", "default", "online").phase == "Running"


def test_qos_handlers_write_contracts():
    store = ObjectStore()
    store.create("Node", synth.make_node("n1"))
    off = synth.make_pod("off", "g", cpu_milli=2000, mem=GI,
                         node_name="n1", phase="Running")
    off.meta.annotations["volcano.sh/preemptable"] = "true"
    store.create("Pod", off)
    mgr = EventsManager(store, "n1")
    mgr.register(CpuQosHandler())
    mgr.register(MemoryQosHandler(factor=1.5))
    mgr.register(NetworkQosHandler(total_bps=10e9, offline_share=0.4))
    mgr.probe.injected = NodeUsage("n1", cpu_pct=85.0, mem_pct=40.0)
    mgr.tick()
    pod = store.get("Pod", "default", "off")
    assert pod.meta.annotations["qos.volcano.sh/cpu-quota-milli"] == "200"
    assert pod.meta.annotations["qos.volcano.sh/memory-high"] == \
        str(int(GI * 1.5))
    node = store.get("Node", "default", "n1")
    # cpu > 80 → offline bandwidth halved: 10e9 * 0.4 * 0.5
    assert node.meta.annotations["qos.volcano.sh/offline-bandwidth-bps"] == \
        str(int(10e9 * 0.2))


def test_agent_scheduler_fast_path():
    store = ObjectStore()
    for n in synth.make_nodes(4, cpu_milli=4000, mem=16 * GI):
        store.create("Node", n)
    binder = FakeBinder()
    asched = AgentScheduler(store, binder=binder, workers=3)
    # bare pods (no podgroup annotation) ride the fast path
    for i in range(6):
        p = synth.make_pod(f"fp-{i}", podgroup="", cpu_milli=1000, mem=GI)
        p.meta.annotations.pop("scheduling.volcano.sh/group-name", None)
        store.create("Pod", p)
    bound = asched.run_once()
    assert bound == 6
    assert len(binder.binds) == 6
    # capacity respected: 4 nodes × 4 cpu = 16; six 1-cpu pods spread
    from collections import Counter
    per_node = Counter(binder.binds.values())
    assert max(per_node.values()) <= 4


def test_sharding_two_phase_handoff():
    store = ObjectStore()
    for n in synth.make_nodes(6):
        store.create("Node", n)
    ctrl = ShardingController(shards=2)
    ctrl.initialize(store)
    ctrl.sync_once()
    s0 = store.get("NodeShard", "default", "shard-0")
    s1 = store.get("NodeShard", "default", "shard-1")
    assert len(s0.nodes_desired) + len(s1.nodes_desired) == 6
    assert s0.nodes_to_add == s0.nodes_desired    # phase 1
    # phase 2: scheduler acknowledges
    owned = ShardingController.acknowledge(store, 0)
    assert owned == s0.nodes_desired
    s0 = store.get("NodeShard", "default", "shard-0")
    assert s0.nodes_to_add == [] and s0.nodes_in_use == owned
    # node removed → toRemove staged for the owner
    gone = s0.nodes_in_use[0]
    store.delete("Node", "default", gone)
    ctrl.sync_once()
    s0 = store.get("NodeShard", "default", "shard-0")
    assert gone in s0.nodes_to_remove or gone not in s0.nodes_desired


def test_colocation_config_projection():
    store = ObjectStore()
    store.create("Node", synth.make_node("pool-a", labels={"pool": "a"}))
    store.create("Node", synth.make_node("pool-b", labels={"pool": "b"}))
    store.create("ColocationConfig", ColocationConfig(
        meta=ObjectMeta(name="cfg-a"), node_selector={"pool": "a"},
        oversubscription_enable=True, oversubscription_ratio=0.7))
    cm = ControllerManager(store, ["colocationconfig"])
    cm.sync_until_quiet()
    a = store.get("Node", "default", "pool-a")
    b = store.get("Node", "default", "pool-b")
    eff = json.loads(a.meta.annotations["colocation.volcano.sh/effective-config"])
    assert eff["oversubscription"] is True and eff["oversubscriptionRatio"] == 0.7
    assert "colocation.volcano.sh/effective-config" not in b.meta.annotations


def test_cache_dumper(tmp_path):
    from volcano_amd.scheduler.daemon import dump_cache
    store = ObjectStore()
    for n in synth.make_nodes(2, cpu_milli=2000, mem=4 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "dj", replicas=2, cpu_milli=500, mem=GI)
    cache = SchedulerCache(store=store, binder=FakeBinder())
    Scheduler(cache).run_once()
    path = str(tmp_path / "dump.json")
    dump_cache(cache, path)
    with open(path) as f:
        data = json.load(f)
    assert len(data["nodes"]) == 2
    assert data["jobs"]["default/dj"]["phase"] == "Running"
    statuses = {t["status"] for t in
                data["jobs"]["default/dj"]["tasks"].values()}
    assert statuses == {"BOUND"}


def test_agent_scheduler_capacity_race():
    """Two workers, one slot: the serialized commit section must admit
    exactly one of the racing pods (optimistic conflict → unschedulable)."""
    store = ObjectStore()
    store.create("Node", synth.make_node("solo", cpu_milli=1000, mem=4 * GI))
    binder = FakeBinder()
    asched = AgentScheduler(store, binder=binder, workers=2)
    for i in range(2):
        p = synth.make_pod(f"race-{i}", podgroup="", cpu_milli=1000, mem=GI)
        p.meta.annotations.pop("scheduling.volcano.sh/group-name", None)
        store.create("Pod", p)
    bound = asched.run_once()
    assert bound == 1
    assert len(binder.binds) == 1
    # node not oversubscribed
    ni = asched.cache.nodes["solo"]
    assert ni.used.milli_cpu <= 1000.5
