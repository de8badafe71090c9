"""Reference reclaim-action table cases through uthelper (reference
actions/reclaim/reclaim_test.go:41-112): cross-queue fairness — an
overusing queue gives back exactly one preemptable pod to a starving
queue, and among overusing queues the lowest-priority job's preemptable
pod is reclaimed first."""

from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

G = 10 ** 9
RECLAIM = ["enqueue", "allocate", "reclaim", "backfill"]


def pod(ns, name, pg_name, cpu, mem, node="", phase="Pending",
        preemptable=None, prio=0):
    p = synth.make_pod(name, pg_name, namespace=ns, cpu_milli=cpu, mem=mem,
                       node_name=node, phase=phase, priority=prio)
    if preemptable is not None:
        p.meta.annotations["volcano.sh/preemptable"] = preemptable
    return p


def pg(ns, name, queue, min_member, phase="Inqueue", prio=0):
    g = synth.make_podgroup(name, queue=queue, namespace=ns,
                            min_member=min_member)
    g.status.phase = phase
    if prio:
        g.meta.annotations["priority"] = str(prio)
    return g


def test_reclaim_from_overusing_queue():
    """reclaim_test.go:50 — q1 holds the whole 3-CPU node (deserved 1.5);
    q2's pending pod reclaims the ONE explicitly-preemptable victim."""
    t = TestCommonStruct(
        name="two queues, one overusing: reclaim",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=1),
                   pg("c1", "pg2", "q2", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=1),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptee3", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=1),
              pod("c1", "preemptor1", "pg2", 1000, G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=3000, mem=3 * G, pods=10)],
        queues=[synth.make_queue("q1"), synth.make_queue("q2")],
        actions=RECLAIM,
        expect_evicted=["c1/preemptee2"],
    ).run()
    t.check_all()


def test_reclaim_sorts_reclaimees_by_job_priority():
    """reclaim_test.go:77 — two overusing queues (mid- and low-priority
    jobs); the LOW-priority job's preemptable pod is reclaimed first,
    and its non-preemptable sibling is protected."""
    t = TestCommonStruct(
        name="sort reclaimees when reclaiming from overusing queue",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=500),
                   pg("c1", "pg2", "q2", 1, prio=100),
                   pg("c1", "pg3", "q3", 1, prio=1000)],
        pods=[pod("c1", "preemptee1-1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=500),
              pod("c1", "preemptee1-2", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=500),
              pod("c1", "preemptee2-1", "pg2", 1000, G, "n1", "Running",
                  preemptable="true", prio=100),
              pod("c1", "preemptee2-2", "pg2", 1000, G, "n1", "Running",
                  preemptable="false", prio=100),
              pod("c1", "preemptor1", "pg3", 1000, G, prio=1000)],
        nodes=[synth.make_node("n1", cpu_milli=4000, mem=4 * G, pods=10)],
        queues=[synth.make_queue("q1"), synth.make_queue("q2"),
                synth.make_queue("q3")],
        actions=RECLAIM,
        expect_evicted=["c1/preemptee2-1"],
    ).run()
    t.check_all()
