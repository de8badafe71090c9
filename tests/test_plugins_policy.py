"""Policy plugin suite: capacity (hierarchical queues), nodegroup, tdm,
usage, pdb, cdp, resource-strategy-fit."""

import time

import pytest

from volcano_amd.api.objects import (ObjectMeta, PodDisruptionBudget, Queue,
                                     QueueSpec)
from volcano_amd.api.resource import CPU, MEMORY, Resource
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import (PluginOption, SchedulerConfiguration,
                                          Tier)
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk(tiers=None, actions=None):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    if tiers is not None:
        config.tiers = tiers
    if actions is not None:
        config.actions = actions
    sched = Scheduler(cache, config)
    return store, binder, cache, sched


def tiers_with(*extra, base=("priority", "gang", "conformance"),
               tier2=("overcommit", "drf", "predicates", "nodeorder",
                      "binpack")):
    t2 = [PluginOption(n) for n in tier2]
    for name, args in extra:
        t2.append(PluginOption(name, arguments=args))
    return [Tier(plugins=[PluginOption(n) for n in base]), Tier(plugins=t2)]


def test_capacity_explicit_deserved():
    tiers = tiers_with(("capacity", {}))
    store, binder, cache, sched = mk(tiers=tiers)
    for n in synth.make_nodes(1, cpu_milli=10000, mem=64 * GI):
        store.create("Node", n)
    qa = synth.make_queue("qa")
    qa.spec.deserved = Resource({CPU: 3000.0})
    qb = synth.make_queue("qb")
    store.create("Queue", qa)
    store.create("Queue", qb)
    synth.make_gang(store, "ja", replicas=8, min_member=1, queue="qa",
                    cpu_milli=1000, mem=GI)
    synth.make_gang(store, "jb", replicas=8, min_member=1, queue="qb",
                    cpu_milli=1000, mem=GI)
    sched.run_once()
    ja = sum(1 for k in binder.binds if k.startswith("default/ja"))
    jb = sum(1 for k in binder.binds if k.startswith("default/jb"))
    assert ja == 3          # capped by explicit deserved 3 cpu
    assert jb == 7          # the remainder


def test_capacity_hierarchy():
    tiers = tiers_with(("capacity", {}))
    store, binder, cache, sched = mk(tiers=tiers)
    for n in synth.make_nodes(1, cpu_milli=12000, mem=64 * GI):
        store.create("Node", n)
    # root splits 12 cpu between team-a (w2) and team-b (w1);
    # team-a's children x (w1) and y (w1) split team-a's 8
    for name, weight, parent in [("team-a", 2, ""), ("team-b", 1, ""),
                                 ("x", 1, "team-a"), ("y", 1, "team-a")]:
        q = synth.make_queue(name, weight=weight)
        q.spec.parent = parent
        store.create("Queue", q)
    synth.make_gang(store, "jx", replicas=12, min_member=1, queue="x",
                    cpu_milli=1000, mem=GI)
    synth.make_gang(store, "jy", replicas=12, min_member=1, queue="y",
                    cpu_milli=1000, mem=GI)
    synth.make_gang(store, "jb", replicas=12, min_member=1, queue="team-b",
                    cpu_milli=1000, mem=GI)
    sched.run_once()
    got = {q: sum(1 for k in binder.binds if k.startswith(f"default/j{q[0]}"))
           for q in ("x", "y", "b")}
    assert got["x"] == 4 and got["y"] == 4    # team-a 8 split 4/4
    assert got["b"] == 4                      # team-b deserved 4


def test_nodegroup_affinity():
    tiers = tiers_with(("nodegroup", {}))
    store, binder, cache, sched = mk(tiers=tiers)
    store.create("Node", synth.make_node(
        "g1-node", cpu_milli=8000, mem=32 * GI,
        labels={"volcano.sh/nodegroup-name": "group1"}))
    store.create("Node", synth.make_node(
        "g2-node", cpu_milli=8000, mem=32 * GI,
        labels={"volcano.sh/nodegroup-name": "group2"}))
    q = synth.make_queue("gq")
    q.spec.affinity = {"affinity": {"required": ["group1"]}}
    store.create("Queue", q)
    synth.make_gang(store, "gj", replicas=2, queue="gq", cpu_milli=1000,
                    mem=GI)
    sched.run_once()
    assert len(binder.binds) == 2
    assert set(binder.binds.values()) == {"g1-node"}


def test_tdm_revocable_window():
    now = time.localtime()
    active = f"{(now.tm_hour - 1) % 24:02d}:00-{(now.tm_hour + 1) % 24:02d}:59"
    closed = f"{(now.tm_hour + 2) % 24:02d}:00-{(now.tm_hour + 3) % 24:02d}:00"
    tiers = tiers_with(("tdm", {"tdm.revocable-zone.z1": active}))
    store, binder, cache, sched = mk(tiers=tiers)
    store.create("Node", synth.make_node(
        "rev", cpu_milli=4000, mem=16 * GI,
        labels={"volcano.sh/revocable-zone": "z1"}))
    store.create("Queue", synth.make_queue("default"))
    # EXPLICITLY non-preemptable job cannot use the revocable node
    # (reference GetPodPreemptable: unset defaults to preemptable=true,
    # so plain pods ARE revocable-eligible — explicit "false" is not)
    pg_np = synth.make_podgroup("np", min_member=1)
    store.create("PodGroup", pg_np)
    pod_np = synth.make_pod("np-w-0", "np", cpu_milli=1000, mem=GI)
    pod_np.meta.annotations["volcano.sh/preemptable"] = "false"
    store.create("Pod", pod_np)
    sched.run_once()
    assert binder.binds == {}
    # default (unset ⇒ preemptable) job can (active window)
    pg = synth.make_podgroup("pp", min_member=1)
    store.create("PodGroup", pg)
    pod = synth.make_pod("pp-w-0", "pp", cpu_milli=1000, mem=GI)
    store.create("Pod", pod)
    sched.run_once()
    assert "default/pp-w-0" in binder.binds


def test_usage_threshold_filters_node():
    tiers = tiers_with(("usage", {"usage.cpu-threshold": 50}))
    store, binder, cache, sched = mk(tiers=tiers)
    busy = synth.make_node("busy", cpu_milli=8000, mem=32 * GI)
    busy.meta.annotations["volcano.sh/cpu-usage"] = "90"
    idle = synth.make_node("idle", cpu_milli=8000, mem=32 * GI)
    idle.meta.annotations["volcano.sh/cpu-usage"] = "10"
    store.create("Node", busy)
    store.create("Node", idle)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "uj", replicas=2, cpu_milli=1000, mem=GI)
    sched.run_once()
    assert set(binder.binds.values()) == {"idle"}


def test_pdb_protects_victims():
    tiers = tiers_with(("pdb", {}))
    store, binder, cache, sched = mk(
        tiers=tiers, actions=["enqueue", "allocate", "preempt", "backfill"])
    for n in synth.make_nodes(1, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    store.create("PodDisruptionBudget", PodDisruptionBudget(
        meta=ObjectMeta(name="pdb1"),
        selector={"volcano.sh/job-name": "low"}, min_available=2))
    synth.make_gang(store, "low", replicas=2, min_member=0, cpu_milli=1000,
                    mem=GI, priority=1)
    sched.run_once()
    synth.make_gang(store, "high", replicas=1, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    # without the pdb both victims are evictable (min_member=0); the pdb
    # pins minAvailable=2 → no eviction possible
    assert binder.evictions == []


def test_cdp_cooldown_protection():
    tiers = tiers_with(("cdp", {"cdp.cooldown-time": "3600s"}))
    store, binder, cache, sched = mk(
        tiers=tiers, actions=["enqueue", "allocate", "preempt", "backfill"])
    for n in synth.make_nodes(1, cpu_milli=1000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "fresh", replicas=1, min_member=0, cpu_milli=1000,
                    mem=GI, priority=1)
    sched.run_once()
    synth.make_gang(store, "hot", replicas=1, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert binder.evictions == []     # victim is inside its cooldown


def test_resource_strategy_fit_directions():
    # cpu MostAllocated (binpack), memory LeastAllocated (spread):
    # with equal cpu usage, the memory-lighter node must win
    tiers = tiers_with(("resource-strategy-fit", {
        "resources": {"cpu": {"type": "MostAllocated", "weight": 1},
                      "memory": {"type": "LeastAllocated", "weight": 5}}}),
        tier2=("overcommit", "drf", "predicates", "proportion"))
    store, binder, cache, sched = mk(tiers=tiers)
    a = synth.make_node("mem-heavy", cpu_milli=8000, mem=32 * GI)
    b = synth.make_node("mem-light", cpu_milli=8000, mem=32 * GI)
    store.create("Node", a)
    store.create("Node", b)
    store.create("Queue", synth.make_queue("default"))
    # pre-load mem-heavy with a memory hog
    synth.make_gang(store, "hog", replicas=1, cpu_milli=100, mem=16 * GI,
                    node_name="")
    sched.run_once()
    hog_node = list(binder.binds.values())[0]
    other = "mem-heavy" if hog_node == "mem-light" else "mem-light"
    synth.make_gang(store, "probe", replicas=1, cpu_milli=100, mem=GI)
    sched.run_once()
    assert binder.binds["default/probe-worker-0"] == other


def test_topology_spread_across_zones():
    tiers = tiers_with(("topologyspread", {}))
    store, binder, cache, sched = mk(tiers=tiers)
    for i, zone in enumerate(["z1", "z1", "z2", "z2"]):
        store.create("Node", synth.make_node(
            f"n{i}", cpu_milli=8000, mem=32 * GI, labels={"zone": zone}))
    store.create("Queue", synth.make_queue("default"))

    def spread_pod(name, pg):
        p = synth.make_pod(name, pg, cpu_milli=500, mem=GI)
        p.affinity = {"topologySpread": {"group": "web", "topologyKey":
                                         "zone", "maxSkew": 1}}
        return p

    zones = {"n0": "z1", "n1": "z1", "n2": "z2", "n3": "z2"}
    placed_zones = []
    # one pod per cycle (exact enforcement granularity)
    for i in range(4):
        pg = synth.make_podgroup(f"sp{i}", min_member=1)
        store.create("PodGroup", pg)
        store.create("Pod", spread_pod(f"sp{i}-w-0", f"sp{i}"))
        sched.run_once()
        node = binder.binds.get(f"default/sp{i}-w-0")
        assert node is not None
        placed_zones.append(zones[node])
    # with maxSkew 1 the four pods alternate: 2 in each zone
    assert placed_zones.count("z1") == 2
    assert placed_zones.count("z2") == 2


def test_rescheduling_high_node_utilization():
    """highNodeUtilization strategy: offline pods on overloaded nodes
    become shuffle victims (reference rescheduling strategy registry)."""
    from volcano_amd.scheduler.config import PluginOption

    store, binder, cache, sched = mk(
        actions=["enqueue", "allocate", "shuffle"])
    sched_cfg = sched.config
    sched_cfg.tiers[1].plugins.append(PluginOption(
        "rescheduling",
        arguments={"interval": 0.0,
                   "strategies": "highNodeUtilization",
                   "highNodeUtilization": {"cpu": 50.0}}))
    import volcano_amd.scheduler.plugins.rescheduling as rs
    rs.ReschedulingPlugin._last_run = 0.0
    for n in synth.make_nodes(1, cpu_milli=4000, mem=16 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("hot", min_member=1)
    store.create("PodGroup", pg)
    pod = synth.make_pod("hot-w-0", "hot", cpu_milli=3000, mem=GI)
    pod.meta.annotations["volcano.sh/preemptable"] = "true"
    store.create("Pod", pod)
    sched.run_once()          # binds; node at 75% cpu > 50% threshold
    rs.ReschedulingPlugin._last_run = 0.0
    sched.run_once()          # shuffle evicts the offline pod
    assert "default/hot-w-0" in binder.evictions
