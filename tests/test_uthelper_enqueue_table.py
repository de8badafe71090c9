"""Reference enqueue-action table cases through uthelper (reference
actions/enqueue/enqueue_test.go:46-126): phase transitions on admission
— Inqueue stays, Pending admits when the queue gates pass, Running is
untouched, a missing queue blocks, and minResources above the queue
capability blocks."""

from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

G = 10 ** 9
ENQ = ["enqueue"]


def pg(ns, name, queue, min_member, phase="Pending", minres=None):
    g = synth.make_podgroup(name, queue=queue, namespace=ns,
                            min_member=min_member, min_resources=minres)
    g.status.phase = phase
    return g


def pod(ns, name, pg_name, cpu, mem, phase="Pending"):
    return synth.make_pod(name, pg_name, namespace=ns, cpu_milli=cpu,
                          mem=mem, phase=phase)


def test_enqueue_inqueue_stays_inqueue():
    t = TestCommonStruct(
        name="when podgroup status is inqueue",
        podgroups=[pg("c1", "pg1", "c1", 2, phase="Inqueue")],
        pods=[pod("c1", "p1", "pg1", 1000, G),
              pod("c1", "p2", "pg1", 1000, G)],
        queues=[synth.make_queue("c1", capability={"cpu": 4000.0,
                                                   "memory": 4.0 * G}),
                synth.make_queue("c2", capability={"cpu": 4000.0,
                                                   "memory": 4.0 * G})],
        actions=ENQ,
        expect_status={"c1/pg1": "Inqueue"},
    ).run()
    t.check_all()


def test_enqueue_pending_admits_both_queues():
    t = TestCommonStruct(
        name="when podgroup status is pending",
        podgroups=[pg("c1", "pg1", "c1", 1),
                   pg("c1", "pg2", "c2", 1)],
        pods=[pod("c1", "p1", "pg1", 3000, G),
              pod("c1", "p2", "pg2", 1000, G)],
        queues=[synth.make_queue("c1", capability={"cpu": 4000.0,
                                                   "memory": 4.0 * G}),
                synth.make_queue("c2", capability={"cpu": 4000.0,
                                                   "memory": 4.0 * G})],
        actions=ENQ,
        expect_status={"c1/pg1": "Inqueue", "c1/pg2": "Inqueue"},
    ).run()
    t.check_all()


def test_enqueue_running_untouched():
    t = TestCommonStruct(
        name="when podgroup status is running",
        podgroups=[pg("c1", "pg1", "c1", 2, phase="Running")],
        pods=[pod("c1", "p1", "pg1", 1000, G, phase="Running"),
              pod("c1", "p2", "pg1", 1000, G, phase="Running")],
        queues=[synth.make_queue("c1", capability={"cpu": 4000.0,
                                                   "memory": 4.0 * G})],
        actions=ENQ,
        expect_status={"c1/pg1": "Running"},
    ).run()
    t.check_all()


def test_enqueue_missing_queue_blocks():
    t = TestCommonStruct(
        name="podgroup names queue c1 but only c2 exists",
        podgroups=[pg("c1", "pg1", "c1", 0)],
        queues=[synth.make_queue("c2", capability={"cpu": 4000.0,
                                                   "memory": 4.0 * G})],
        actions=ENQ,
        expect_status={"c1/pg1": "Pending"},
    ).run()
    t.check_all()


def test_enqueue_minresources_above_capability_blocks():
    t = TestCommonStruct(
        name="queue resources less than podgroup MinResources",
        podgroups=[pg("c1", "pg1", "c1", 1,
                      minres={"cpu": 8000.0, "memory": 8.0 * G})],
        queues=[synth.make_queue("c1", capability={"cpu": 1000.0,
                                                   "memory": 1.0 * G})],
        actions=ENQ,
        expect_status={"c1/pg1": "Pending"},
    ).run()
    t.check_all()
