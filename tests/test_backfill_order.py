"""Backfill ordering (reference actions/backfill/backfill_test.go): when
pod slots are scarce, higher-priority jobs' best-effort tasks backfill
first; resourceful pending tasks are not backfill candidates."""

from volcano_amd.api.types import TaskStatus
from volcano_amd.utils import synth
from volcano_amd.utils.uthelper import TestCommonStruct

G = 10 ** 9


def pg(name, prio):
    g = synth.make_podgroup(name, queue="q1", min_member=0)
    g.status.phase = "Inqueue"
    g.meta.annotations["priority"] = str(prio)
    return g


def be_pod(name, pg_name, prio):
    return synth.make_pod(name, pg_name, cpu_milli=0, mem=0, priority=prio)


def test_backfill_prefers_high_priority_job():
    """Two pod slots, four best-effort pods across two jobs — the
    high-priority job's pods take both slots."""
    t = TestCommonStruct(
        name="backfill priority order",
        podgroups=[pg("pg1", 1), pg("pg2", 2)],
        pods=[be_pod("pg1-t1", "pg1", 1), be_pod("pg1-t2", "pg1", 3),
              be_pod("pg2-t1", "pg2", 1), be_pod("pg2-t2", "pg2", 3)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=2)],
        queues=[synth.make_queue("q1")],
        actions=["enqueue", "backfill"],
        expect_bind_count=2,
    ).run()
    t.check_all()
    bound = sorted(t.binder.binds)
    assert all(k.startswith("default/pg2-") for k in bound), bound


def test_backfill_skips_resourceful_pending():
    """A pending task WITH resources is allocate's job, not backfill's."""
    t = TestCommonStruct(
        name="backfill skips unbesteffort",
        podgroups=[pg("pg1", 1)],
        pods=[synth.make_pod("fat", "pg1", cpu_milli=1000, mem=G),
              be_pod("thin", "pg1", 1)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=5)],
        queues=[synth.make_queue("q1")],
        actions=["enqueue", "backfill"],
        expect_bind_count=1,
    ).run()
    t.check_all()
    assert list(t.binder.binds) == ["default/thin"]


def test_allocate_skips_besteffort_backfill_places_it():
    """Reference allocate.go:265: allocate never places BestEffort
    tasks; the same cycle's backfill does."""
    t = TestCommonStruct(
        name="allocate skips best-effort",
        podgroups=[pg("pg1", 1)],
        pods=[synth.make_pod("fat", "pg1", cpu_milli=1000, mem=G),
              be_pod("thin", "pg1", 1)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=5)],
        queues=[synth.make_queue("q1")],
        actions=["enqueue", "allocate"],
        expect_bind_count=1,       # only the resourceful task
    ).run()
    t.check_all()
    assert list(t.binder.binds) == ["default/fat"]

    t2 = TestCommonStruct(
        name="backfill completes the job",
        podgroups=[pg("pg1", 1)],
        pods=[synth.make_pod("fat", "pg1", cpu_milli=1000, mem=G),
              be_pod("thin", "pg1", 1)],
        nodes=[synth.make_node("n1", cpu_milli=12000, mem=12 * G, pods=5)],
        queues=[synth.make_queue("q1")],
        actions=["enqueue", "allocate", "backfill"],
        expect_bind_count=2,
    ).run()
    t2.check_all()
