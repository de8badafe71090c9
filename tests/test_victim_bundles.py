"""Victim-bundle construction semantics (reference
``actions/utils/bundle_test.go``): safe/whole split above job+role
minimums, disjoint task sets, allowWholeBundle selection, and the
no-role-minimum unconstrained case."""

import numpy as np

from volcano_amd.api.info import JobInfo, TaskInfo
from volcano_amd.scheduler.actions.gangpreempt import GangPreemptAction
from volcano_amd.utils import synth

G = 10 ** 9


class StubNT:
    r = 1

    def req_vector(self, t):
        return np.array([t.request.q.get("cpu", 0.0)], dtype=np.float64)


class StubSsn:
    preemptable_fns = []
    reclaimable_fns = []
    node_tensors = StubNT()


def make_victim_job(n_tasks, min_member, role_min=None, prio=1):
    pg = synth.make_podgroup("victim", namespace="c1",
                             min_member=min_member)
    if role_min:
        pg.spec.min_task_member = dict(role_min)
    pg.meta.annotations["priority"] = str(prio)
    job = JobInfo("c1/victim", pg)
    for i in range(n_tasks):
        pod = synth.make_pod(f"v-{i}", "victim", namespace="c1",
                             cpu_milli=1000, mem=G, node_name="n1",
                             phase="Running", role="worker", priority=i)
        job.add_task(TaskInfo.from_pod(pod, job.key))
    return job


def make_preemptor(prio=100):
    pg = synth.make_podgroup("preemptor", namespace="c1", min_member=1)
    pg.meta.annotations["priority"] = str(prio)
    job = JobInfo("c1/preemptor", pg)
    pod = synth.make_pod("p-0", "preemptor", namespace="c1",
                         cpu_milli=1000, mem=G, priority=prio)
    t = TaskInfo.from_pod(pod, job.key)
    job.add_task(t)
    return job, t


def test_create_bundles_safe_and_whole():
    """bundle_test.go:27 — 5 running tasks, min 3, role min worker:3 →
    safe bundle of 2 (lowest priority) + whole bundle of the other 3,
    disjoint."""
    vj = make_victim_job(5, 3, role_min={"worker": 3})
    pj, rep = make_preemptor()
    a = GangPreemptAction()
    bundles = a._victim_bundles(StubSsn(), rep, pj, vj)
    kinds = [b[0] for b in bundles]
    assert kinds == ["safe", "whole"], kinds
    safe, whole = bundles
    assert len(safe[2]) == 2 and len(whole[2]) == 3
    # lowest-priority tasks evict first; sets are disjoint
    assert sorted(t.priority for t in safe[2]) == [0, 1]
    assert not ({t.uid for t in safe[2]} & {t.uid for t in whole[2]})
    assert safe[3][0] == 2000.0 and whole[3][0] == 3000.0


def test_allow_whole_bundle_false_drops_whole():
    """bundle_test.go:81 SelectBundles(allowWhole=false) analog."""
    vj = make_victim_job(5, 3, role_min={"worker": 3})
    pj, rep = make_preemptor()
    a = GangPreemptAction()
    a.allow_whole_bundle = False
    bundles = a._victim_bundles(StubSsn(), rep, pj, vj)
    assert [b[0] for b in bundles] == ["safe"]


def test_no_role_min_unconstrained():
    """bundle_test.go:412 — without TaskMinAvailable the role never caps
    the safe split; only the job minimum does."""
    vj = make_victim_job(5, 2)
    pj, rep = make_preemptor()
    a = GangPreemptAction()
    bundles = a._victim_bundles(StubSsn(), rep, pj, vj)
    safe = next(b for b in bundles if b[0] == "safe")
    assert len(safe[2]) == 3     # surplus above min_member=2


def test_equal_priority_victim_yields_no_bundles():
    """Victims must be strictly below the preemptor's priority."""
    vj = make_victim_job(3, 1, prio=100)
    vj.podgroup.meta.annotations["priority"] = "100"
    pj, rep = make_preemptor(prio=100)
    a = GangPreemptAction()
    assert a._victim_bundles(StubSsn(), rep, pj, vj) == []


def test_bundle_invariants_property():
    """Property sweep: for random victim jobs, the safe bundle never
    dips the gang below job or role minimums, and safe/whole partition
    the victims disjointly."""
    from hypothesis import given, settings
    from hypothesis import strategies as st

    @settings(max_examples=60, deadline=None)
    @given(n=st.integers(1, 12), mm=st.integers(0, 12),
           role_min=st.integers(0, 12), prio=st.integers(0, 5))
    def check(n, mm, role_min, prio):
        vj = make_victim_job(n, mm,
                             role_min={"worker": role_min} if role_min
                             else None, prio=prio)
        pj, rep = make_preemptor(prio=100)
        a = GangPreemptAction()
        bundles = a._victim_bundles(StubSsn(), rep, pj, vj)
        safe = [b for b in bundles if b[0] == "safe"]
        whole = [b for b in bundles if b[0] == "whole"]
        assert len(safe) <= 1 and len(whole) <= 1
        s_ids = {t.uid for b in safe for t in b[2]}
        w_ids = {t.uid for b in whole for t in b[2]}
        assert not (s_ids & w_ids)
        assert len(s_ids) + len(w_ids) == n
        # safe eviction keeps the gang at/above both minimums
        assert n - len(s_ids) >= min(mm, n)
        if role_min:
            assert n - len(s_ids) >= min(role_min, n)

    check()
