"""SubGroupPolicy — gang-within-gang (reference scheduling/v1beta1
types.go:218 SubGroupPolicySpec + allocate's subjob machinery): pods
partition into gang-atomic subgroups of subGroupSize; minSubGroups gates
the whole family; per-subgroup networkTopology confines each subgroup to
its own domain."""

from volcano_amd.api.objects import (HyperNode, HyperNodeMember,
                                     MemberSelector, ObjectMeta)
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def world(n_nodes=4, cpu=4000, topo=False):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    if topo:
        config.tiers[1].plugins.append(
            PluginOption("network-topology-aware"))
    sched = Scheduler(cache, config)
    for i in range(n_nodes):
        store.create("Node", synth.make_node(
            f"n{i}", cpu_milli=cpu, mem=32 * GI))
    store.create("Queue", synth.make_queue("default"))
    return store, binder, cache, sched


def mk_subgroup_job(store, name, replicas, policy, cpu_milli=1000):
    pg = synth.make_podgroup(name, min_member=1)
    pg.spec.sub_group_policy = policy
    store.create("PodGroup", pg)
    for i in range(replicas):
        store.create("Pod", synth.make_pod(
            f"{name}-w-{i}", name, cpu_milli=cpu_milli, mem=GI,
            role="worker"))


def test_subgroups_complete_chunks_only():
    store, binder, cache, sched = world(n_nodes=4, cpu=8000)
    mk_subgroup_job(store, "sg", replicas=5,
                    policy=[{"subGroupSize": 2, "minSubGroups": 0}])
    sched.run_once()
    # 2 complete pairs place; the tail pod has no complete subgroup
    assert len(binder.binds) == 4


def test_min_subgroups_family_gate():
    # capacity for only ONE pair, but the family demands two → nothing
    store, binder, cache, sched = world(n_nodes=2, cpu=1000)
    mk_subgroup_job(store, "fam", replicas=4,
                    policy=[{"subGroupSize": 2, "minSubGroups": 2}])
    sched.run_once()
    assert binder.binds == {}
    # grow the cluster → both pairs fit → all 4 place
    for i in (2, 3):
        store.create("Node", synth.make_node(f"n{i}", cpu_milli=1000,
                                             mem=32 * GI))
    sched.run_once()
    assert len(binder.binds) == 4


def test_subgroup_atomicity():
    # 3 nodes × 1 slot: one full pair + a single → pair must not split
    store, binder, cache, sched = world(n_nodes=3, cpu=1000)
    mk_subgroup_job(store, "at", replicas=4,
                    policy=[{"subGroupSize": 2, "minSubGroups": 0}])
    sched.run_once()
    # only one complete pair fits (2 of 3 slots); never a lone half-pair
    assert len(binder.binds) in (2,)


def test_per_subgroup_topology_domains():
    """Each subgroup confined to ONE rack, different subgroups may take
    different racks — the capability job-level topology cannot express."""
    store, binder, cache, sched = world(n_nodes=4, cpu=2000, topo=True)

    def hn(name, tier, nodes=None, children=None):
        members = []
        if nodes:
            members.append(HyperNodeMember(
                type="Node", selector=MemberSelector(exact_match=nodes)))
        if children:
            members.append(HyperNodeMember(
                type="HyperNode",
                selector=MemberSelector(exact_match=children)))
        return HyperNode(meta=ObjectMeta(name=name), tier=tier,
                         members=members)

    store.create("HyperNode", hn("rack-a", 1, nodes=["n0", "n1"]))
    store.create("HyperNode", hn("rack-b", 1, nodes=["n2", "n3"]))
    store.create("HyperNode", hn("spine", 2,
                                 children=["rack-a", "rack-b"]))
    # 2 subgroups of 2 pods à 2000m: one rack (2×2000m) holds exactly one
    mk_subgroup_job(store, "tp", replicas=4, cpu_milli=2000,
                    policy=[{"subGroupSize": 2, "minSubGroups": 2,
                             "networkTopology": {"mode": "hard",
                                                 "highestTierAllowed": 1}}])
    sched.run_once()
    assert len(binder.binds) == 4
    racks = {"n0": "a", "n1": "a", "n2": "b", "n3": "b"}
    by_sg = {}
    for key, node in binder.binds.items():
        idx = int(key.rsplit("-", 1)[1]) // 2     # pods 0,1 | 2,3 pair up
        by_sg.setdefault(idx, set()).add(racks[node])
    # both racks used, and no subgroup straddles racks is implied by
    # capacity (each rack fits exactly one subgroup)
    used_racks = {r for s in by_sg.values() for r in s}
    assert used_racks == {"a", "b"}


def test_match_label_keys_grouping():
    store, binder, cache, sched = world(n_nodes=4, cpu=8000)
    pg = synth.make_podgroup("mlk", min_member=1)
    pg.spec.sub_group_policy = [{"subGroupSize": 2, "minSubGroups": 0,
                                 "matchLabelKeys": ["shard"]}]
    store.create("PodGroup", pg)
    for i in range(4):
        p = synth.make_pod(f"mlk-w-{i}", "mlk", cpu_milli=500, mem=GI)
        p.meta.labels["shard"] = "s0" if i < 2 else "s1"
        # 3rd pod of s1 never forms a pair with s0's pods
        store.create("Pod", p)
    p = synth.make_pod("mlk-w-4", "mlk", cpu_milli=500, mem=GI)
    p.meta.labels["shard"] = "s1"
    store.create("Pod", p)
    sched.run_once()
    # s0 pair + one s1 pair place; s1's odd pod stays pending
    assert len(binder.binds) == 4


def test_subgroups_with_full_pipeline_churn():
    """Subgroup jobs under the full corrective pipeline across cycles:
    gang atomicity of subgroups holds, accounting stays consistent."""
    import random
    from volcano_amd.api.resource import CPU
    from volcano_amd.api.info import _OCC_SET
    from volcano_amd.scheduler.config import PluginOption as PO
    rng = random.Random(11)
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    config.actions = ["enqueue", "allocate", "preempt", "backfill"]
    sched = Scheduler(cache, config)
    for i in range(8):
        store.create("Node", synth.make_node(f"n{i}", cpu_milli=4000,
                                             mem=32 * GI))
    store.create("Queue", synth.make_queue("default"))
    jid = 0
    for cycle in range(12):
        for _ in range(rng.randint(1, 3)):
            jid += 1
            size = rng.choice([2, 3])
            mk_subgroup_job(store, f"s{jid:03d}",
                            replicas=size * rng.randint(1, 3) +
                            rng.randint(0, 1),
                            policy=[{"subGroupSize": size,
                                     "minSubGroups": 1}],
                            cpu_milli=rng.choice([500, 1000]))
        bound = [p for p in store.list("Pod") if p.node_name]
        for p in rng.sample(bound, min(2, len(bound))):
            store.delete("Pod", p.meta.namespace, p.meta.name)
        sched.run_once()
        # invariants: node accounting consistent; subgroup-size multiples
        for ni in cache.nodes.values():
            rec = sum(t.request.get(CPU) for t in ni.tasks.values()
                      if t.status in _OCC_SET)
            assert abs(rec - ni.used.get(CPU)) < 1.0
        for job in cache.jobs.values():
            if not (job.podgroup and job.podgroup.spec.sub_group_policy):
                continue
            size = job.podgroup.spec.sub_group_policy[0]["subGroupSize"]
            assert job.occupied_count % size == 0, \
                f"cycle {cycle}: {job.key} occupied {job.occupied_count} " \
                f"not a multiple of subgroup size {size}"
    assert len(binder.binds) > 10
