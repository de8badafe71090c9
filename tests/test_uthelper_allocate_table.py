"""Reference allocate-action table cases, ported through the uthelper
harness (reference actions/allocate/allocate_test.go:86-232 — the same
scenarios and expectations, tensor decision plane instead of per-task
Go callbacks)."""

import pytest

from volcano_amd.api.resource import Resource
from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

GI = 1024 ** 3
G = 10 ** 9


def pod(ns, name, pg, cpu, mem, role=None, selector=None):
    p = synth.make_pod(name, pg, namespace=ns, cpu_milli=cpu, mem=mem,
                       role=role or "", node_selector=selector)
    return p


def node(name, cpu, mem, labels=None):
    return synth.make_node(name, cpu_milli=cpu, mem=mem, pods=10,
                           labels=labels)


def pg(ns, name, queue, min_member, mtm=None, phase="Inqueue"):
    g = synth.make_podgroup(name, queue=queue, namespace=ns,
                            min_member=min_member, min_task_member=mtm)
    g.status.phase = phase
    return g


def test_prepredicate_failed_node_selector():
    t = TestCommonStruct(
        name="prepredicate failed: node selector does not match",
        podgroups=[pg("c1", "pg1", "c1", 1)],
        pods=[pod("c1", "p1", "pg1", 1000, G, role="master",
                  selector={"nodeRole": "master"}),
              pod("c1", "p2", "pg1", 1000, G, role="worker",
                  selector={"nodeRole": "worker"})],
        nodes=[node("n1", 2000, 4 * GI, labels={"nodeRole": "worker"})],
        queues=[synth.make_queue("c1")],
        actions=["allocate"],
        expect_bind_map={"c1/p2": "n1"},
        expect_bind_count=1,
    ).run()
    t.check_all()


def test_continue_until_min_member_met():
    t = TestCommonStruct(
        name="prepredicate failed and tasks are not used up",
        podgroups=[pg("c1", "pg1", "c1", 2,
                      mtm={"master": 1, "worker": 1})],
        pods=[pod("c1", "p0", "pg1", 1000, G, role="master",
                  selector={"nodeRole": "master"}),
              pod("c1", "p1", "pg1", 1000, G, role="master",
                  selector={"nodeRole": "master"}),
              pod("c1", "p2", "pg1", 1000, G, role="worker",
                  selector={"nodeRole": "worker"}),
              pod("c1", "p3", "pg1", 1000, G, role="worker",
                  selector={"nodeRole": "worker"})],
        nodes=[node("n1", 1000, 2 * GI, labels={"nodeRole": "master"}),
               node("n2", 1000, 2 * GI, labels={"nodeRole": "worker"})],
        queues=[synth.make_queue("c1")],
        actions=["allocate"],
        expect_bind_map={"c1/p0": "n1", "c1/p2": "n2"},
        expect_bind_count=2,
    ).run()
    t.check_all()


def test_master_min_member_unmet_breaks_allocation():
    t = TestCommonStruct(
        name="master's min member can not be allocated",
        podgroups=[pg("c1", "pg1", "c1", 2,
                      mtm={"master": 2, "worker": 0})],
        pods=[pod("c1", "p0", "pg1", 1000, G, role="master",
                  selector={"nodeRole": "master"}),
              pod("c1", "p1", "pg1", 1000, G, role="master",
                  selector={"nodeRole": "master"}),
              pod("c1", "p2", "pg1", 1000, G, role="worker",
                  selector={"nodeRole": "worker"}),
              pod("c1", "p3", "pg1", 1000, G, role="worker",
                  selector={"nodeRole": "worker"})],
        nodes=[node("n1", 1000, 2 * GI, labels={"nodeRole": "master"}),
               node("n2", 2000, 2 * GI, labels={"nodeRole": "worker"})],
        queues=[synth.make_queue("c1")],
        actions=["allocate"],
        expect_bind_map={},
        expect_bind_count=0,
    ).run()
    t.check_all()


def test_one_job_two_pods_one_node():
    t = TestCommonStruct(
        name="one Job with two Pods on one node",
        podgroups=[pg("c1", "pg1", "c1", 0)],
        pods=[pod("c1", "p1", "pg1", 1000, G),
              pod("c1", "p2", "pg1", 1000, G)],
        nodes=[node("n1", 2000, 4 * GI)],
        queues=[synth.make_queue("c1")],
        actions=["allocate"],
        expect_bind_map={"c1/p1": "n1", "c1/p2": "n1"},
        expect_bind_count=2,
    ).run()
    t.check_all()


def test_two_jobs_fair_share_one_node():
    """Two queues, one 2-cpu node, 1-cpu pods: proportion gives each
    queue one placement (the reference's DRF/proportion fairness
    outcome)."""
    t = TestCommonStruct(
        name="two Jobs on one node",
        podgroups=[pg("c1", "pg1", "c1", 0), pg("c2", "pg2", "c2", 0)],
        pods=[pod("c1", "pg1-p-1", "pg1", 1000, G),
              pod("c1", "pg1-p-2", "pg1", 1000, G),
              pod("c2", "pg2-p-1", "pg2", 1000, G),
              pod("c2", "pg2-p-2", "pg2", 1000, G)],
        nodes=[node("n1", 2000, 4 * GI)],
        queues=[synth.make_queue("c1"), synth.make_queue("c2")],
        actions=["allocate"],
        expect_bind_map={"c1/pg1-p-1": "n1", "c2/pg2-p-1": "n1"},
        expect_bind_count=2,
    ).run()
    t.check_all()


def test_big_job_does_not_block_other_queue():
    t = TestCommonStruct(
        name="high priority queue should not block others",
        podgroups=[pg("c1", "pg1", "c1", 0), pg("c1", "pg2", "c2", 0)],
        pods=[pod("c1", "p1", "pg1", 3000, G),
              pod("c1", "p2", "pg2", 1000, G)],
        nodes=[node("n1", 2000, 4 * GI)],
        queues=[synth.make_queue("c1"), synth.make_queue("c2")],
        actions=["allocate"],
        expect_bind_map={"c1/p2": "n1"},
        expect_bind_count=1,
    ).run()
    t.check_all()
