"""Reference shuffle-action table case through uthelper (reference
actions/shuffle/shuffle_test.go:49-121): a fake victim plugin selects
every low-priority running pod; shuffle evicts exactly those across
jobs and nodes."""

from volcano_amd.scheduler.plugins.base import Plugin, register
from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

G = 10 ** 9
LOW, HIGH = 1, 100


@register("fake-lowprio-victims")
class FakeLowPrioVictims(Plugin):
    """shuffle_test.go:94-107 fakePluginVictimFns analog."""

    def on_session_open(self, ssn) -> None:
        ssn.victim_tasks_fns.append(
            lambda candidates: [t for t in candidates
                                if t.priority == LOW])


def pod(name, pg_name, node, prio):
    return synth.make_pod(name, pg_name, namespace="test", cpu_milli=1000,
                          mem=2 * G, node_name=node, phase="Running",
                          priority=prio)


def pg(name):
    g = synth.make_podgroup(name, queue="default", namespace="test",
                            min_member=0)
    g.status.phase = "Running"
    return g


def test_shuffle_evicts_low_priority_pods():
    t = TestCommonStruct(
        name="select pods with low priority and evict them",
        nodes=[synth.make_node("node1", cpu_milli=4000, mem=8 * G, pods=10),
               synth.make_node("node2", cpu_milli=4000, mem=8 * G, pods=10)],
        queues=[synth.make_queue("default")],
        podgroups=[pg("pg1"), pg("pg2"), pg("pg3")],
        pods=[pod("pod1-1", "pg1", "node1", LOW),
              pod("pod1-2", "pg1", "node1", HIGH),
              pod("pod1-3", "pg1", "node1", HIGH),
              pod("pod2-1", "pg2", "node1", LOW),
              pod("pod2-2", "pg2", "node2", HIGH),
              pod("pod3-1", "pg3", "node2", LOW),
              pod("pod3-2", "pg3", "node2", HIGH)],
        tiers=[["fake-lowprio-victims"]],
        actions=["shuffle"],
        expect_evicted=["test/pod1-1", "test/pod2-1", "test/pod3-1"],
    ).run()
    t.check_all()
