"""Reference preempt-action table cases through uthelper (reference
actions/preempt/preempt_test.go:109-240): idle-resources short-circuit,
pipelined short-circuit, evict-one, evict-just-enough, queue-capability
pressure, and the explicit preemptable=false protection (#2232)."""

from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

GI = 1024 ** 3
G = 10 ** 9
PREEMPT = ["enqueue", "allocate", "preempt", "backfill"]


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


def q(name, cap=None):
    return synth.make_queue(name, capability=cap)


def test_no_preempt_when_idle_resources():
    t = TestCommonStruct(
        name="do not preempt if there are enough idle resources",
        podgroups=[pg("c1", "pg1", "q1", 3)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running"),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running"),
              pod("c1", "preemptor1", "pg1", 1000, G)],
        nodes=[synth.make_node("n1", cpu_milli=10000, mem=10 * G, pods=10)],
        queues=[q("q1")],
        actions=PREEMPT,
        expect_evicted=[],
    ).run()
    t.check_all()


def test_no_preempt_when_job_pipelined():
    t = TestCommonStruct(
        name="do not preempt if job is pipelined",
        podgroups=[pg("c1", "pg1", "q1", 1), pg("c1", "pg2", "q1", 1)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running"),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running"),
              pod("c1", "preemptee3", "pg2", 1000, G, "n1", "Running"),
              pod("c1", "preemptor2", "pg2", 1000, G)],
        nodes=[synth.make_node("n1", cpu_milli=3000, mem=3 * G, pods=10)],
        queues=[q("q1")],
        actions=PREEMPT,
        expect_evicted=[],
    ).run()
    t.check_all()


def test_preempt_one_task_of_lower_priority_job():
    t = TestCommonStruct(
        name="preempt one task of different job to fit both jobs",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=1),
              pod("c1", "preemptor1", "pg2", 1000, G, prio=100),
              pod("c1", "preemptor2", "pg2", 1000, G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=2000, mem=2 * G, pods=10)],
        queues=[q("q1")],
        actions=PREEMPT,
        expect_evicted=["c1/preemptee1"],
    ).run()
    t.check_all()


def test_preempt_enough_tasks_for_large_preemptor():
    t = TestCommonStruct(
        name="preempt enough tasks to fit large task",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptee3", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=1),
              pod("c1", "preemptor1", "pg2", 5000, 5 * G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=6000, mem=6 * G, pods=10)],
        queues=[q("q1")],
        actions=PREEMPT,
        expect_evicted=["c1/preemptee1", "c1/preemptee2"],
    ).run()
    t.check_all()


def test_preempt_under_queue_capability_pressure():
    """#3161: queue at capability — the low-prio job's task must go."""
    t = TestCommonStruct(
        name="preempt low priority job in same queue",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 3000, 3 * G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptor1", "pg2", 3000, 3 * G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=10)],
        queues=[q("q1", cap={"cpu": 4000.0, "memory": 4.0 * G})],
        actions=PREEMPT,
        expect_evicted=["c1/preemptee1"],
    ).run()
    t.check_all()


def test_no_preempt_when_queue_has_headroom():
    """#3161 inverse: capability 6 — allocate places the preemptor
    without any eviction."""
    t = TestCommonStruct(
        name="allocatable and has enough resource, don't preempt",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 3000, 3 * G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptor1", "pg2", 3000, 3 * G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=10)],
        queues=[q("q1", cap={"cpu": 6000.0, "memory": 6.0 * G})],
        actions=PREEMPT,
        expect_evicted=[],
        expect_bind_count=1,
    ).run()
    t.check_all()


def test_preempt_protects_nonpreemptable_and_higher_priority_pods():
    """preempt_test.go: queue at capability 3 — only the explicitly
    preemptable, lower-priority victim goes; preemptable=false and
    equal/higher pod priority are both protected."""
    t = TestCommonStruct(
        name="not pod with preemptable=false or higher priority",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="false", prio=1),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptee3", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=100),
              pod("c1", "preemptor1", "pg2", 1000, G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=10)],
        queues=[q("q1", cap={"cpu": 3000.0, "memory": 3.0 * G})],
        actions=PREEMPT,
        expect_evicted=["c1/preemptee2"],
    ).run()
    t.check_all()


def test_unbesteffort_preempts_besteffort_pod_slot():
    """#3335: the node has ONE pod slot held by a best-effort pod; the
    resourceful high-priority preemptor evicts it on the pods dim."""
    t = TestCommonStruct(
        name="unBestEffort preempts BestEffort",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 0, 0, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptor1", "pg2", 3000, 3 * G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=1)],
        queues=[q("q1", cap={"cpu": 6000.0, "memory": 6.0 * G})],
        actions=PREEMPT,
        expect_evicted=["c1/preemptee1"],
    ).run()
    t.check_all()


def test_besteffort_preempts_besteffort_pod_slot():
    """#3335: both best-effort — still contend on the pods dim."""
    t = TestCommonStruct(
        name="BestEffort preempts BestEffort",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 0, 0, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptor1", "pg2", 0, 0, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=1)],
        queues=[q("q1", cap={"cpu": 6000.0, "memory": 6.0 * G})],
        actions=PREEMPT,
        expect_evicted=["c1/preemptee1"],
    ).run()
    t.check_all()


def test_preemption_policy_never_blocks_preemption():
    """#3642 (preempt.go:441): a preemptor with preemptionPolicy=Never
    waits instead of displacing the running pod."""
    never = pod("c1", "preemptor1", "pg2", 1000, G, prio=100)
    never.preemption_policy = "Never"
    t = TestCommonStruct(
        name="task preemption policy never",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              never],
        nodes=[synth.make_node("n1", cpu_milli=1000, mem=G, pods=1)],
        queues=[q("q1")],
        actions=PREEMPT,
        expect_evicted=[],
    ).run()
    t.check_all()


def test_preempt_rollback_when_queue_capacity_fails():
    """preempt_test.go "rollback evictions...": the node has idle room
    and evictable victims, but the preemptor (4 CPU) exceeds the queue's
    3-CPU capability even after evicting — no eviction may commit."""
    t = TestCommonStruct(
        name="rollback when preemptor exceeds queue capacity",
        podgroups=[pg("c1", "pg1", "q1", 1, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptee2", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptor1", "pg2", 4000, 4 * G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=10000, mem=10 * G, pods=10)],
        queues=[q("q1", cap={"cpu": 3000.0, "memory": 3.0 * G})],
        actions=PREEMPT,
        expect_evicted=[],
    ).run()
    t.check_all()


def test_preempt_commits_only_on_winning_node():
    """preempt_test.go "only commit evictions on the node where
    preemption succeeds": n1's small victim cannot bring the queue
    under its capability, n2's large one can — only n2's eviction
    commits."""
    t = TestCommonStruct(
        name="only commit on winning node",
        podgroups=[pg("c1", "pg1", "q1", 0, prio=1),
                   pg("c1", "pg2", "q1", 1, prio=100),
                   pg("c1", "pg3", "q1", 0, prio=1)],
        pods=[pod("c1", "preemptee1", "pg1", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptee2", "pg3", 4000, 4 * G, "n2", "Running",
                  preemptable="true", prio=1),
              pod("c1", "preemptor1", "pg2", 3000, 3 * G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=3000, mem=3 * G, pods=10),
               synth.make_node("n2", cpu_milli=4000, mem=4 * G, pods=10)],
        queues=[q("q1", cap={"cpu": 6000.0, "memory": 6.0 * G})],
        actions=PREEMPT,
        expect_evicted=["c1/preemptee2"],
    ).run()
    t.check_all()


def test_preempt_multi_queue_independent():
    """preempt_test.go multi-queue regression: a non-starving q1 must
    not interfere with q2's preemptor — pg3 evicts pg2's pod regardless
    of queue iteration order."""
    t = TestCommonStruct(
        name="multi-queue preemptor isolation",
        podgroups=[pg("c1", "pg1", "q1", 1),
                   pg("c1", "pg2", "q2", 0, prio=1),
                   pg("c1", "pg3", "q2", 1, prio=100)],
        pods=[pod("c1", "q1-runner1", "pg1", 1000, G, "n1", "Running"),
              pod("c1", "q2-preemptee1", "pg2", 1000, G, "n1", "Running",
                  preemptable="true", prio=1),
              pod("c1", "q2-preemptor1", "pg3", 1000, G, prio=100)],
        nodes=[synth.make_node("n1", cpu_milli=2000, mem=2 * G, pods=10)],
        queues=[q("q1"),
                q("q2", cap={"cpu": 4000.0, "memory": 4.0 * G})],
        actions=PREEMPT,
        expect_evicted=["c1/q2-preemptee1"],
    ).run()
    t.check_all()
