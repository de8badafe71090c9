"""Hierarchical capacity reclaim with ancestor levels (reference
``plugins/capacity/capacity_test.go`` cases 1/4/5/6/7/8 —
``ancestorReclaimLevel`` semantics: leaf deserved checks, parent/
grandparent gates, shared-ancestor no-contention skip)."""

from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

G = 1024 ** 3
GPU = "amd.com/gpu"
RECLAIM = ["enqueue", "allocate", "reclaim", "backfill"]


def run_case(level, queues, podgroups, pods, nodes, expect_evicted,
             expect_pipelined=None):
    t = TestCommonStruct(
        queues=queues, podgroups=podgroups, pods=pods, nodes=nodes,
        tiers=[["capacity", "gang", "predicates", "priority"]],
        plugin_args={"capacity": {"ancestorReclaimLevel": level},
                     "gang": {"enabledReclaimable": False}},
        actions=RECLAIM,
        expect_evicted=expect_evicted,
        expect_pipelined=expect_pipelined,
    ).run()
    t.check_all()
    return t


def pg(name, queue, phase, mm=1):
    g = synth.make_podgroup(name, queue=queue, namespace="ns1",
                            min_member=mm)
    g.status.phase = phase
    return g


def level_tree():
    """capacity_test.go buildLevelSemanticsTree: two branches with
    deserved tightening toward the leaves."""
    return [
        synth.make_queue("root"),
        synth.make_queue("grand-a", parent="root",
                         deserved={"cpu": 2000.0, "memory": 2.0 * G},
                         capability={"cpu": 4000.0, "memory": 4.0 * G}),
        synth.make_queue("parent-a", parent="grand-a",
                         deserved={"cpu": 2000.0, "memory": 2.0 * G},
                         capability={"cpu": 2000.0, "memory": 2.0 * G}),
        synth.make_queue("queue-a", parent="parent-a",
                         deserved={"cpu": 2000.0, "memory": 2.0 * G},
                         capability={"cpu": 2000.0, "memory": 2.0 * G}),
        synth.make_queue("grand-b", parent="root",
                         deserved={"cpu": 3000.0, "memory": 3.0 * G},
                         capability={"cpu": 4000.0, "memory": 4.0 * G}),
        synth.make_queue("parent-b", parent="grand-b",
                         deserved={"cpu": 1000.0, "memory": 1.0 * G},
                         capability={"cpu": 2000.0, "memory": 2.0 * G}),
        synth.make_queue("queue-b", parent="parent-b",
                         deserved={"cpu": 1000.0, "memory": 1.0 * G},
                         capability={"cpu": 2000.0, "memory": 2.0 * G}),
    ]


def level_pods():
    return [synth.make_pod("p-victim", "pg-victim", namespace="ns1",
                           cpu_milli=2000, mem=2 * G, node_name="n1",
                           phase="Running"),
            synth.make_pod("p-reclaimer", "pg-reclaimer", namespace="ns1",
                           cpu_milli=2000, mem=2 * G)]


def level_pgs():
    return [pg("pg-victim", "queue-b", "Running"),
            pg("pg-reclaimer", "queue-a", "Inqueue")]


def level_node():
    return [synth.make_node("n1", cpu_milli=2000, mem=2 * G, pods=10)]


def test_case5_level0_leaf_deserved_allows_reclaim():
    """queue-b holds 2 CPU against deserved 1 → leaf check evicts."""
    run_case(0, level_tree(), level_pgs(), level_pods(), level_node(),
             expect_evicted=["ns1/p-victim"], expect_pipelined=1)


def test_case6_level1_parent_gate_passes():
    """parent-b subtree (2 CPU) exceeds its deserved 1 → still evicts."""
    run_case(1, level_tree(), level_pgs(), level_pods(), level_node(),
             expect_evicted=["ns1/p-victim"], expect_pipelined=1)


def test_case7_level2_grandparent_gate_blocks():
    """grand-b subtree (2 CPU) is UNDER its deserved 3 → blocked."""
    run_case(2, level_tree(), level_pgs(), level_pods(), level_node(),
             expect_evicted=[])


def test_case8_shared_parent_unset_leaf_deserved_blocks():
    """Leaves without deserved under one shared parent: the
    no-real-contention skip protects the sibling's pod."""
    queues = [
        synth.make_queue("root"),
        synth.make_queue("parent", parent="root",
                         deserved={"cpu": 4000.0, "memory": 4.0 * G},
                         capability={"cpu": 4000.0, "memory": 4.0 * G}),
        synth.make_queue("queue-a", parent="parent"),
        synth.make_queue("queue-b", parent="parent"),
    ]
    run_case(1, queues, level_pgs(), level_pods(), level_node(),
             expect_evicted=[])


def test_case1_parent_deserved_reclaim():
    """capacity_test.go case1: the reclaimer's branch carries a parent
    deserved (1 GPU); the victim queue's branch has none — the GPU hog
    is reclaimed at level 1."""
    queues = [
        synth.make_queue("root",
                         deserved={"cpu": 16000.0, "memory": 16.0 * G,
                                   GPU: 4.0}),
        synth.make_queue("case1_queue1", parent="root",
                         deserved={GPU: 1.0}),
        synth.make_queue("case1_queue11", parent="case1_queue1"),
        synth.make_queue("case1_queue2", parent="root"),
    ]
    podgroups = [pg("pg1", "case1_queue2", "Running"),
                 pg("pg2", "case1_queue11", "Inqueue")]
    pods = [synth.make_pod("p1", "pg1", namespace="ns1", cpu_milli=1000,
                           mem=G, node_name="n1", phase="Running",
                           extra={GPU: 4.0}),
            synth.make_pod("p2", "pg2", namespace="ns1", cpu_milli=1000,
                           mem=G, extra={GPU: 1.0})]
    nodes = [synth.make_node("n1", cpu_milli=16000, mem=16 * G, pods=11,
                             extra={GPU: 4.0})]
    run_case(1, queues, podgroups, pods, nodes,
             expect_evicted=["ns1/p1"], expect_pipelined=1)


def test_case4_victim_child_within_deserved_blocks():
    """capacity_test.go case4: the victim's leaf holds exactly its
    deserved GPU — cross-parent reclaim must not evict it."""
    queues = [
        synth.make_queue("root",
                         deserved={"cpu": 16000.0, "memory": 16.0 * G,
                                   GPU: 4.0}),
        synth.make_queue("case4_parent1", parent="root",
                         deserved={GPU: 1.0}),
        synth.make_queue("case4_child1", parent="case4_parent1",
                         deserved={GPU: 1.0}),
        synth.make_queue("case4_child1_sibling", parent="case4_parent1"),
        synth.make_queue("case4_parent2", parent="root",
                         deserved={GPU: 1.0}),
        synth.make_queue("case4_child2", parent="case4_parent2",
                         deserved={GPU: 1.0}),
    ]
    podgroups = [pg("pg1", "case4_child1", "Running"),
                 pg("pg2", "case4_child2", "Pending"),
                 pg("pg3", "case4_child1_sibling", "Running")]
    p3 = synth.make_pod("p3", "pg3", namespace="ns1", cpu_milli=1000,
                        mem=G, node_name="n1", phase="Running",
                        extra={GPU: 1.0})
    p3.meta.annotations["volcano.sh/preemptable"] = "false"
    pods = [synth.make_pod("p1", "pg1", namespace="ns1", cpu_milli=1000,
                           mem=G, node_name="n1", phase="Running",
                           extra={GPU: 1.0}),
            synth.make_pod("p2", "pg2", namespace="ns1", cpu_milli=1000,
                           mem=G, extra={GPU: 1.0}),
            p3]
    nodes = [synth.make_node("n1", cpu_milli=8000, mem=8 * G, pods=11,
                             extra={GPU: 1.0})]
    run_case(1, queues, podgroups, pods, nodes, expect_evicted=[])
