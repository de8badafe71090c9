"""HyperNode tree + network-topology-aware scheduling."""

from volcano_amd.api.hypernode import HyperNodeTree
from volcano_amd.api.objects import (HyperNode, HyperNodeMember,
                                     MemberSelector, ObjectMeta)
from volcano_amd.controllers import ControllerManager
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption, Tier
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk_hn(name, tier, nodes=None, children=None):
    members = []
    if nodes:
        members.append(HyperNodeMember(
            type="Node", selector=MemberSelector(exact_match=nodes)))
    if children:
        members.append(HyperNodeMember(
            type="HyperNode", selector=MemberSelector(exact_match=children)))
    return HyperNode(meta=ObjectMeta(name=name), tier=tier, members=members)


def test_tree_membership_and_lca():
    hns = [
        mk_hn("rack-a", 1, nodes=["n1", "n2"]),
        mk_hn("rack-b", 1, nodes=["n3", "n4"]),
        mk_hn("spine", 2, children=["rack-a", "rack-b"]),
    ]
    tree = HyperNodeTree(hns, ["n1", "n2", "n3", "n4"])
    assert tree.members["spine"] == {"n1", "n2", "n3", "n4"}
    assert tree.leaf_of["n1"] == "rack-a"
    assert tree.lca_tier("n1", "n2") == 1
    assert tree.lca_tier("n1", "n3") == 2


def test_tree_regex_members():
    hns = [mk_hn("rack-a", 1)]
    hns[0].members = [HyperNodeMember(
        type="Node", selector=MemberSelector(regex_match=r"gpu-\d+"))]
    tree = HyperNodeTree(hns, ["gpu-1", "gpu-2", "cpu-1"])
    assert tree.members["rack-a"] == {"gpu-1", "gpu-2"}


def topo_world():
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    config.tiers[1].plugins.append(PluginOption("network-topology-aware"))
    sched = Scheduler(cache, config)
    # two racks of 2 nodes, 4 cpu each
    for i, rack in [(0, "a"), (1, "a"), (2, "b"), (3, "b")]:
        store.create("Node", synth.make_node(f"n{i}", cpu_milli=4000,
                                             mem=16 * GI))
    store.create("HyperNode", mk_hn("rack-a", 1, nodes=["n0", "n1"]))
    store.create("HyperNode", mk_hn("rack-b", 1, nodes=["n2", "n3"]))
    store.create("HyperNode", mk_hn("spine", 2,
                                    children=["rack-a", "rack-b"]))
    store.create("Queue", synth.make_queue("default"))
    return store, binder, cache, sched


def test_hard_topology_confines_job():
    store, binder, cache, sched = topo_world()
    pg = synth.make_podgroup("tj", min_member=2)
    pg.spec.network_topology = {"mode": "hard", "highestTierAllowed": 1}
    store.create("PodGroup", pg)
    for i in range(2):
        store.create("Pod", synth.make_pod(f"tj-w-{i}", "tj",
                                           cpu_milli=2000, mem=GI))
    sched.run_once()
    assert len(binder.binds) == 2
    nodes = set(binder.binds.values())
    # both pods inside ONE rack
    assert nodes <= {"n0", "n1"} or nodes <= {"n2", "n3"}


def test_hard_topology_infeasible_blocks():
    store, binder, cache, sched = topo_world()
    # needs 10 cpu > any single rack's 8 → tier-1 hard constraint fails
    pg = synth.make_podgroup("big", min_member=5)
    pg.spec.network_topology = {"mode": "hard", "highestTierAllowed": 1}
    store.create("PodGroup", pg)
    for i in range(5):
        store.create("Pod", synth.make_pod(f"big-w-{i}", "big",
                                           cpu_milli=2000, mem=GI))
    sched.run_once()
    assert binder.binds == {}
    # tier 2 allowed → spine domain fits it
    pg.spec.network_topology = {"mode": "hard", "highestTierAllowed": 2}
    store.update("PodGroup", pg)
    sched.run_once()
    assert len(binder.binds) == 5


def test_soft_topology_falls_back():
    store, binder, cache, sched = topo_world()
    pg = synth.make_podgroup("soft", min_member=5)
    pg.spec.network_topology = {"mode": "soft", "highestTierAllowed": 1}
    store.create("PodGroup", pg)
    for i in range(5):
        store.create("Pod", synth.make_pod(f"soft-w-{i}", "soft",
                                           cpu_milli=2000, mem=GI))
    sched.run_once()
    assert len(binder.binds) == 5      # soft: unconstrained fallback


def test_hypernode_controller_label_discovery():
    store = ObjectStore()
    for i in range(4):
        store.create("Node", synth.make_node(
            f"n{i}", labels={"topology.volcano.sh/rack": f"r{i // 2}",
                             "topology.volcano.sh/spine": "s0"}))
    cm = ControllerManager(store, ["hypernode"])
    cm.sync_until_quiet()
    hns = {h.meta.name: h for h in store.list("HyperNode")}
    assert set(hns) == {"rack-r0", "rack-r1", "spine-s0"}
    assert hns["rack-r0"].tier == 1
    assert hns["spine-s0"].tier == 2
    tree = HyperNodeTree(list(hns.values()), [f"n{i}" for i in range(4)])
    assert tree.members["spine-s0"] == {"n0", "n1", "n2", "n3"}


def test_hypernode_controller_fabric_discovery(tmp_path):
    """UFM-dump provider: leaf/spine HyperNodes from interface records
    (reference discovery/ufm/ufm.go buildHyperNodes)."""
    import json
    dump = [
        {"system_name": "lsw1", "tier": 1, "node_description": "n0"},
        {"system_name": "lsw1", "tier": 1, "node_description": "n1"},
        {"system_name": "lsw2", "tier": 1, "node_description": "n2"},
        {"system_name": "sp0", "tier": 2, "peer_node_name": "lsw1"},
        {"system_name": "sp0", "tier": 2, "peer_node_name": "lsw2"},
    ]
    path = tmp_path / "fabric.json"
    path.write_text(json.dumps(dump))
    store = ObjectStore()
    cm = ControllerManager(store, ["hypernode"])
    for c in cm.controllers:
        if c.name == "hypernode":
            c.fabric_file = str(path)
    cm.sync_until_quiet()
    hns = {h.meta.name: h for h in store.list("HyperNode")}
    assert set(hns) == {"leaf-lsw1", "leaf-lsw2", "spine-sp0"}
    tree = HyperNodeTree(list(hns.values()), ["n0", "n1", "n2"])
    assert tree.members["spine-sp0"] == {"n0", "n1", "n2"}
    assert tree.lca_tier("n0", "n1") == 1
    assert tree.lca_tier("n0", "n2") == 2


def test_topology_aware_preemption_into_domain():
    """a hard-topology gang whose domains are all
    occupied by lower-priority work preempts its way INTO one domain via
    dry-run trials (reference preempt.go:479 topologyAwarePreempt), and
    the pipelined tasks carry node nominations across cycles."""
    store, binder, cache, sched = topo_world()
    sched2 = None
    # add preempt to the action list
    from volcano_amd.scheduler.config import default_config as dc
    config = dc()
    config.actions = ["enqueue", "allocate", "preempt", "backfill"]
    config.tiers[1].plugins.append(PluginOption("network-topology-aware"))
    sched = Scheduler(cache, config)

    # low-prio filler occupies both racks (2 cpu free per node: 4-node
    # cluster, each node 4 cpu, filler takes 3 cpu/node)
    synth.make_gang(store, "filler-a", replicas=4, min_member=1,
                    cpu_milli=3000, mem=GI, priority=1)
    sched.run_once()
    assert binder.bound_count == 4
    # bound tasks are preemptable victims (ALLOCATED_STATUSES)

    # hard-topology gang needs 2x3 cpu in ONE rack — infeasible without
    # eviction (each node has only 1 cpu free)
    pg = synth.make_podgroup("net", min_member=2)
    pg.spec.network_topology = {"mode": "hard", "highestTierAllowed": 1}
    store.create("PodGroup", pg)
    for i in range(2):
        store.create("Pod", synth.make_pod(
            f"net-w-{i}", "net", cpu_milli=3000, mem=GI, priority=100))
    sched.run_once()

    job = cache.jobs["default/net"]
    # the gang is pipelined into exactly one rack with victims evicted
    assert len(binder.evictions) >= 1
    pipelined = job.tasks_with_status(
        __import__("volcano_amd.api.types", fromlist=["TaskStatus"])
        .TaskStatus.PIPELINED)
    assert len(pipelined) == 2
    nodes = {t.node_name for t in pipelined}
    assert nodes <= {"n0", "n1"} or nodes <= {"n2", "n3"}, nodes

    # evicted filler pods terminate; the nomination fast path (or the
    # normal scored path) binds the gang into the SAME domain
    for key in list(binder.evictions):
        ns, name = key.split("/")
        pod = store.get("Pod", ns, name)
        if pod is not None:
            store.delete("Pod", ns, name)
    for _ in range(3):
        sched.run_once()
        got = {k: v for k, v in binder.binds.items() if k.startswith("default/net")}
        if len(got) == 2:
            break
    assert len(got) == 2, binder.binds
    bound_nodes = set(got.values())
    assert bound_nodes <= {"n0", "n1"} or bound_nodes <= {"n2", "n3"}
