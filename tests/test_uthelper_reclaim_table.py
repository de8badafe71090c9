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


def test_reclaim_prefers_lowest_priority_queue():
    """reclaim_test.go:113 — overusing queues with different queue
    priority: the LOWEST-priority queue's preemptable pod goes first."""
    t = TestCommonStruct(
        name="sort reclaimees by queue priority",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=500),
                   pg("c1", "pg2", "q2", 1, prio=500),
                   pg("c1", "pg3", "q3", 1, prio=500)],
        pods=[pod("c1", "preemptee1-1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=500),
              pod("c1", "preemptee1-2", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=500),
              pod("c1", "preemptee2-1", "pg2", 1000, G, "n1", "Running",
                  preemptable="true", prio=500),
              pod("c1", "preemptee2-2", "pg2", 1000, G, "n1", "Running",
                  preemptable="false", prio=500),
              pod("c1", "preemptor1", "pg3", 1000, G, prio=500)],
        nodes=[synth.make_node("n1", cpu_milli=4000, mem=4 * G, pods=10)],
        queues=[synth.make_queue("q1", priority=5),
                synth.make_queue("q2", priority=10),
                synth.make_queue("q3", priority=1)],
        actions=RECLAIM,
        expect_evicted=["c1/preemptee1-1"],
    ).run()
    t.check_all()


def test_reclaim_blocked_by_preemption_policy_never():
    """reclaim_test.go:144 (#3642): a reclaimer pod with
    preemptionPolicy=Never waits instead of reclaiming."""
    never = pod("c1", "preemptor1", "pg2", 1000, G, prio=1000)
    never.preemption_policy = "Never"
    t = TestCommonStruct(
        name="reclaim blocked by PreemptNever",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=100),
                   pg("c1", "pg2", "q2", 0, prio=1000)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=100),
              never],
        nodes=[synth.make_node("n1", cpu_milli=1000, mem=G, pods=1)],
        queues=[synth.make_queue("q1", weight=5),
                synth.make_queue("q2", weight=10)],
        actions=RECLAIM,
        expect_evicted=[],
    ).run()
    t.check_all()


def _closed_queue_case(tiers):
    t = TestCommonStruct(
        name="closed queue cannot reclaim",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=100, phase="Running"),
                   pg("c1", "pg2", "q2", 1, prio=1000)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=100),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=100),
              pod("c1", "preemptee3", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=100),
              pod("c1", "preemptor1", "pg2", 1000, G, prio=1000)],
        nodes=[synth.make_node("n1", cpu_milli=3000, mem=3 * G, pods=10)],
        queues=[synth.make_queue("q1"),
                synth.make_queue("q2", state="Closed")],
        actions=RECLAIM,
        tiers=tiers,
        expect_evicted=[],
    ).run()
    t.check_all()


def test_reclaim_closed_queue_proportion():
    """reclaim_test.go:173 — reclaimer's queue Closed (proportion)."""
    _closed_queue_case(None)


def test_reclaim_closed_queue_capacity():
    """reclaim_test.go:200 — same under the capacity plugin."""
    _closed_queue_case([["conformance", "gang", "capacity"]])


def test_reclaim_counts_node_idle_toward_need():
    """reclaim_test.go:227 — the node has idle CPU but is 1G short on
    memory: evict exactly ONE pod (from the lowest-priority queue), not
    two."""
    t = TestCommonStruct(
        name="node idle counts toward reclaimed resources",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=500, phase="Running"),
                   pg("c1", "pg2", "q2", 0, prio=500, phase="Running"),
                   pg("c1", "pg3", "q3", 1, prio=500)],
        pods=[pod("c1", "preemptee1-1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=500),
              pod("c1", "preemptee2-1", "pg2", 1000, G, "n1", "Running",
                  preemptable="true", prio=500),
              pod("c1", "preemptor1", "pg3", 2000, G, prio=500)],
        nodes=[synth.make_node("n1", cpu_milli=10000, mem=2 * G, pods=10)],
        queues=[synth.make_queue("q1", priority=1),
                synth.make_queue("q2", priority=2),
                synth.make_queue("q3", priority=3)],
        actions=RECLAIM,
        expect_evicted=["c1/preemptee1-1"],
    ).run()
    t.check_all()


def test_reclaim_second_task_when_first_is_preempt_never():
    """reclaim_test.go:262 — per-TASK eligibility: the Never-policy task
    waits, but its gang-mate without the policy still reclaims."""
    never = pod("c1", "preemptor-task1", "pg2", 1000, G, prio=1000)
    never.preemption_policy = "Never"
    t = TestCommonStruct(
        name="second task reclaims when first is PreemptNever",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=100, phase="Running"),
                   pg("c1", "pg2", "q2", 2, prio=1000)],
        pods=[pod("c1", "victim-pod-no", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=100),
              pod("c1", "victim-pod", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=100),
              never,
              pod("c1", "preemptor-task2", "pg2", 1000, G, prio=900)],
        nodes=[synth.make_node("n1", cpu_milli=2000, mem=2 * G, pods=10)],
        queues=[synth.make_queue("q1"), synth.make_queue("q2")],
        actions=RECLAIM,
        tiers=[["conformance", "gang", "proportion", "priority"]],
        plugin_args={"gang": {"enabledJobPipelined": False}},
        expect_evicted=["c1/victim-pod"],
    ).run()
    t.check_all()


def test_reclaim_no_eviction_when_node_idle_suffices():
    """reclaim_test.go:312 regression — FutureIdle already fits the
    preemptor: zero evictions committed even though the victim's queue
    is over its deserved share."""
    t = TestCommonStruct(
        name="no eviction when idle suffices",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=100, phase="Running"),
                   pg("c1", "pg2", "q2", 1, prio=100)],
        pods=[pod("c1", "victim-n1", "pg1", 2000, 2 * G, "n1", "Running",
                  preemptable="true", prio=100),
              pod("c1", "preemptor1", "pg2", 3000, 3 * G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=10000, mem=10 * G, pods=10)],
        queues=[synth.make_queue("q1", weight=1),
                synth.make_queue("q2", weight=9)],
        actions=RECLAIM,
        expect_evicted=[],
        expect_pipelined=None,
    ).run()
    t.check_all()
