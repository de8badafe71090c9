"""Admission breadth: table cases ported from the reference's
admit_job_test.go."""

import pytest

from volcano_amd.api.objects import (Job, JobSpec, LifecyclePolicy,
                                     ObjectMeta, Queue, QueueSpec, TaskSpec)
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.webhooks import AdmissionError, default_chain


def mk_store(queues=("default",)):
    store = ObjectStore()
    for q in queues:
        store.create("Queue", synth.make_queue(q))
    return store


def mk_job(name="j", tasks=None, **spec_kw):
    tasks = tasks if tasks is not None else [
        TaskSpec(name="task-1", replicas=2,
                 template={"resources": {"cpu": "1"}})]
    return Job(meta=ObjectMeta(name=name, namespace="default"),
               spec=JobSpec(tasks=tasks, **spec_kw))


CASES = [
    # (case name from admit_job_test.go, job kwargs, expect error substr)
    ("simple-valid-job", {}, None),
    ("simple-valid-job-with-policy",
     {"policies": [LifecyclePolicy(event="PodEvicted",
                                   action="RestartJob")]}, None),
    ("duplicate-task-job",
     {"tasks": [TaskSpec(name="duplicated-task-1", replicas=1),
                TaskSpec(name="duplicated-task-1", replicas=1)]},
     "duplicated task"),
    ("job-policy-duplicated",
     {"policies": [LifecyclePolicy(event="PodFailed", action="AbortJob"),
                   LifecyclePolicy(event="PodFailed",
                                   action="RestartJob")]},
     "duplicated policy event"),
    ("min-available-illegal", {"min_available": 5}, "total replicas"),
    ("min-available-negative", {"min_available": -1}, ">= 0"),
    ("job-plugin-illegal", {"plugins": {"big_plugin": []}},
     "unable to find job plugin"),
    ("mpi-master-missing",
     {"plugins": {"mpi": ["master=mpimaster"]},
      "tasks": [TaskSpec(name="worker", replicas=2)]},
     "mpi master task was not found"),
    ("mpi-master-present",
     {"plugins": {"mpi": ["master=mpimaster"]},
      "tasks": [TaskSpec(name="mpimaster", replicas=1),
                TaskSpec(name="worker", replicas=2)]}, None),
    ("no-task", {"tasks": []}, "No task specified"),
    ("task-name-not-dns", {"tasks": [TaskSpec(name="Bad_Name",
                                              replicas=1)]}, "DNS-1123"),
    ("job-name-not-dns", None, "DNS-1123"),           # special-cased below
    ("policy-star-exclusive",
     {"policies": [LifecyclePolicy(event="*", action="AbortJob"),
                   LifecyclePolicy(event="PodFailed",
                                   action="RestartJob")]},
     "'*' policy"),
    ("invalid-policy-event",
     {"policies": [LifecyclePolicy(event="NotAnEvent",
                                   action="AbortJob")]},
     "invalid policy event"),
    ("invalid-policy-action",
     {"policies": [LifecyclePolicy(event="PodFailed",
                                   action="FlyAway")]},
     "invalid policy action"),
    ("negative-ttl", {"ttl_seconds_after_finished": -5.0},
     "ttlSecondsAfterFinished"),
    ("task-dependson-cycle",
     {"tasks": [TaskSpec(name="a", replicas=1, depends_on=["b"]),
                TaskSpec(name="b", replicas=1, depends_on=["a"])]},
     "DAG"),
    ("network-topology-bad-mode",
     {"network_topology": {"mode": "diagonal"}}, "hard|soft"),
    ("network-topology-bad-tier",
     {"network_topology": {"mode": "hard", "highestTierAllowed": 0}},
     ">= 1"),
]


@pytest.mark.parametrize("name,kw,expect", CASES,
                         ids=[c[0] for c in CASES])
def test_job_validation_table(name, kw, expect):
    store = mk_store()
    chain = default_chain(store)
    if name == "job-name-not-dns":
        job = mk_job(name="Invalid_Job_Name")
    else:
        job = mk_job(**kw)
    if expect is None:
        chain.admit("Job", job, "CREATE")
    else:
        with pytest.raises(AdmissionError, match=expect):
            chain.admit("Job", job, "CREATE")


def test_submit_to_closed_or_missing_queue():
    store = mk_store()
    chain = default_chain(store)
    with pytest.raises(AdmissionError, match="does not exist"):
        chain.admit("Job", mk_job(queue="nope"), "CREATE")


def test_leaf_queue_only_submission():
    store = mk_store(("default", "parent", "child"))
    child = store.get("Queue", "default", "child")
    child.spec.parent = "parent"
    store.update("Queue", child)
    chain = default_chain(store)
    with pytest.raises(AdmissionError, match="leaf queue"):
        chain.admit("Job", mk_job(queue="parent"), "CREATE")
    chain.admit("Job", mk_job(queue="child"), "CREATE")   # leaf OK


def test_root_queue_rejected():
    store = mk_store(("default", "root"))
    chain = default_chain(store)
    with pytest.raises(AdmissionError, match="root queue"):
        chain.admit("Job", mk_job(queue="root"), "CREATE")


def test_queue_validate_depth():
    """Reference validate_queue_test.go: state legality, guarantee ≤
    deserved ≤ capability, hierarchy/weights alignment, and delete
    protection for default/root and parents with children."""
    import pytest

    from volcano_amd.utils import synth
    from volcano_amd.webhooks import AdmissionError, default_chain
    from volcano_amd.store import ObjectStore

    store = ObjectStore()
    chain = default_chain(store)
    g = chain.guard(store)
    G = 1024 ** 3

    # state legality
    bad = synth.make_queue("q-badstate", state="abnormal-case")
    with pytest.raises(AdmissionError, match="state"):
        g.create("Queue", bad)

    # deserved >= guarantee required (deserved missing)
    with pytest.raises(AdmissionError, match="deserved"):
        g.create("Queue", synth.make_queue("q-g", guarantee={"cpu": 2000.0}))

    # capability >= deserved
    with pytest.raises(AdmissionError, match="capability"):
        g.create("Queue", synth.make_queue(
            "q-cd", capability={"cpu": 1000.0}, deserved={"cpu": 2000.0}))

    # legal tree
    g.create("Queue", synth.make_queue(
        "q-ok", capability={"cpu": 4000.0}, deserved={"cpu": 2000.0},
        guarantee={"cpu": 1000.0}))

    # hierarchy / weights mismatch
    hq = synth.make_queue("q-h")
    hq.meta.annotations["volcano.sh/hierarchy"] = "root/sci"
    hq.meta.annotations["volcano.sh/hierarchy-weights"] = "100/50/25"
    with pytest.raises(AdmissionError, match="depth"):
        g.create("Queue", hq)
    hq.meta.annotations["volcano.sh/hierarchy-weights"] = "100/-1"
    with pytest.raises(AdmissionError, match="positive"):
        g.create("Queue", hq)
    hq.meta.annotations["volcano.sh/hierarchy-weights"] = "100/50"
    g.create("Queue", hq)

    # delete protection
    g.create("Queue", synth.make_queue("default"))
    with pytest.raises(AdmissionError, match="can not be deleted"):
        g.delete("Queue", "default", "default")
    g.create("Queue", synth.make_queue("parent-q"))
    g.create("Queue", synth.make_queue("child-q", parent="parent-q"))
    with pytest.raises(AdmissionError, match="child queues"):
        g.delete("Queue", "default", "parent-q")
    g.delete("Queue", "default", "child-q")
    g.delete("Queue", "default", "parent-q")      # drains bottom-up


def test_pod_validate_jdb_annotations():
    """Reference admit_pod.go: jdb-min-available / jdb-max-unavailable
    must be positive int-or-percentage and mutually exclusive."""
    import pytest

    from volcano_amd.utils import synth
    from volcano_amd.store import ObjectStore
    from volcano_amd.webhooks import AdmissionError, default_chain

    store = ObjectStore()
    g = default_chain(store).guard(store)
    g.create("Queue", synth.make_queue("default"))

    ok = synth.make_pod("p-ok", "")
    ok.meta.annotations["scheduling.volcano.sh/jdb-min-available"] = "30%"
    g.create("Pod", ok)

    bad = synth.make_pod("p-bad", "")
    bad.meta.annotations["scheduling.volcano.sh/jdb-min-available"] = "zero"
    with pytest.raises(AdmissionError, match="positive integer"):
        g.create("Pod", bad)

    both = synth.make_pod("p-both", "")
    both.meta.annotations["scheduling.volcano.sh/jdb-min-available"] = "1"
    both.meta.annotations["scheduling.volcano.sh/jdb-max-unavailable"] = "1"
    with pytest.raises(AdmissionError, match="multiple annotations"):
        g.create("Pod", both)


def test_podgroup_inherits_namespace_queue():
    """Reference mutate_podgroup_test.go: a podgroup on the default
    queue picks up its namespace's queue-name annotation; an explicit
    queue wins."""
    from volcano_amd.api.objects import Namespace, ObjectMeta
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    from volcano_amd.webhooks import default_chain

    store = ObjectStore()
    g = default_chain(store).guard(store)
    ns = Namespace(meta=ObjectMeta(name="team-ns", namespace=""))
    ns.meta.annotations["scheduling.volcano.sh/queue-name"] = "ns-queue"
    store.create("Namespace", ns)

    pg = synth.make_podgroup("pg-default", namespace="team-ns")
    g.create("PodGroup", pg)
    assert pg.spec.queue == "ns-queue"

    pg2 = synth.make_podgroup("pg-explicit", namespace="team-ns",
                              queue="custom-queue")
    g.create("PodGroup", pg2)
    assert pg2.spec.queue == "custom-queue"

    pg3 = synth.make_podgroup("pg-plain", namespace="other-ns")
    g.create("PodGroup", pg3)
    assert pg3.spec.queue == "default"
