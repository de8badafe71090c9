"""Admission webhooks, apiserver REST surface, vcctl CLI."""

import pytest

from volcano_amd.api.objects import (CronJob, FlowStep, Job, JobFlow, JobSpec,
                                     LifecyclePolicy, ObjectMeta, Queue,
                                     QueueSpec, TaskSpec)
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.webhooks import AdmissionError, default_chain


def chain_store():
    store = ObjectStore()
    store.create("Queue", synth.make_queue("default"))
    chain = default_chain(store)
    return store, chain


def mk_job(name="j", tasks=None, **kw):
    return Job(meta=ObjectMeta(name=name),
               spec=JobSpec(tasks=tasks or [
                   TaskSpec(name="w", replicas=2,
                            template={"resources": {"cpu": "1"}})], **kw))


def test_job_mutate_defaults():
    store, chain = chain_store()
    job = mk_job(tasks=[TaskSpec(name="", replicas=3,
                                 template={"resources": {"cpu": "1"}})])
    job.spec.queue = ""
    chain.admit("Job", job)
    assert job.spec.queue == "default"
    assert job.spec.tasks[0].name == "task-0"
    assert job.spec.min_available == 3


def test_job_validate_rejects():
    store, chain = chain_store()
    with pytest.raises(AdmissionError, match="duplicated task names"):
        chain.admit("Job", mk_job(tasks=[
            TaskSpec(name="a", replicas=1, template={}),
            TaskSpec(name="a", replicas=1, template={})]))
    with pytest.raises(AdmissionError, match="minAvailable"):
        chain.admit("Job", mk_job(min_available=5))
    with pytest.raises(AdmissionError, match="does not exist"):
        chain.admit("Job", mk_job(queue="nope"))
    with pytest.raises(AdmissionError, match="invalid policy action"):
        chain.admit("Job", mk_job(policies=[
            LifecyclePolicy(events=["PodFailed"], action="Explode")]))


def test_queue_hierarchy_validation():
    store, chain = chain_store()
    with pytest.raises(AdmissionError, match="parent queue"):
        chain.admit("Queue", Queue(meta=ObjectMeta(name="child"),
                                   spec=QueueSpec(parent="ghost")))


def test_jobflow_validation():
    store, chain = chain_store()
    with pytest.raises(AdmissionError, match="cycle"):
        chain.admit("JobFlow", JobFlow(
            meta=ObjectMeta(name="bad"),
            flows=[FlowStep(name="a", depends_on=["b"]),
                   FlowStep(name="b", depends_on=["a"])]))
    with pytest.raises(AdmissionError, match="unknown dependency"):
        chain.admit("JobFlow", JobFlow(
            meta=ObjectMeta(name="bad2"),
            flows=[FlowStep(name="a", depends_on=["zzz"])]))


def test_cronjob_validation():
    store, chain = chain_store()
    with pytest.raises(AdmissionError, match="invalid cron"):
        chain.admit("CronJob", CronJob(meta=ObjectMeta(name="c"),
                                       schedule="nonsense"))
    chain.admit("CronJob", CronJob(meta=ObjectMeta(name="c"),
                                   schedule="*/5 * * * *"))


def test_apiserver_rest_roundtrip():
    from fastapi.testclient import TestClient
    from volcano_amd.api.objects import to_dict
    from volcano_amd.store.apiserver import create_app

    store = ObjectStore()
    store.create("Queue", synth.make_queue("default"))
    client = TestClient(create_app(store))

    r = client.get("/healthz")
    assert r.status_code == 200
    job = mk_job("rest-job")
    r = client.post("/apis/Job", json=to_dict(job))
    assert r.status_code == 200, r.text
    r = client.get("/apis/Job/default/rest-job")
    assert r.json()["spec"]["tasks"][0]["replicas"] == 2
    # admission rejection over HTTP
    bad = mk_job("bad", min_available=99)
    r = client.post("/apis/Job", json=to_dict(bad))
    assert r.status_code == 400
    # watch journal
    r = client.get("/watch", params={"since": 0, "kinds": "Job"})
    evs = r.json()["events"]
    assert any(e["object"]["meta"]["name"] == "rest-job" for e in evs)
    # delete
    assert client.delete("/apis/Job/default/rest-job").status_code == 200
    assert client.get("/apis/Job/default/rest-job").status_code == 404


def test_vcctl_flow(tmp_path, capsys):
    from volcano_amd.cli.vcctl import main

    state = str(tmp_path / "state.json")
    assert main(["--state", state, "queue", "create", "-N", "prod",
                 "-w", "4"]) == 0
    assert main(["--state", state, "job", "run", "-N", "train",
                 "-r", "3", "-q", "prod", "--cpu", "2"]) == 0
    assert main(["--state", state, "job", "list"]) == 0
    out = capsys.readouterr().out
    assert "train" in out
    assert main(["--state", state, "job", "view", "-N", "train"]) == 0
    out = capsys.readouterr().out
    assert "replicas: 3" in out
    assert main(["--state", state, "queue", "list"]) == 0
    out = capsys.readouterr().out
    assert "prod" in out
    # suspend issues a Command the job controller consumes
    assert main(["--state", state, "job", "suspend", "-N", "train"]) == 0
    store = ObjectStore.load(state)
    cmds = store.list("Command")
    assert len(cmds) == 1 and cmds[0].action == "AbortJob"
    assert main(["--state", state, "job", "delete", "-N", "train"]) == 0
    store = ObjectStore.load(state)
    assert store.get("Job", "default", "train") is None


def test_hypernode_webhook_validation():
    from volcano_amd.api.objects import (HyperNode, HyperNodeMember,
                                         MemberSelector, ObjectMeta)
    from volcano_amd.store import ObjectStore
    from volcano_amd.webhooks import AdmissionError, default_chain
    import pytest
    store = ObjectStore()
    guarded = default_chain(store).guard(store)
    good = HyperNode(meta=ObjectMeta(name="ok"), tier=1, members=[
        HyperNodeMember(type="Node",
                        selector=MemberSelector(regex_match=r"gpu-\d+"))])
    guarded.create("HyperNode", good)
    bad_re = HyperNode(meta=ObjectMeta(name="badre"), tier=1, members=[
        HyperNodeMember(type="Node",
                        selector=MemberSelector(regex_match="gpu-("))])
    with pytest.raises(AdmissionError, match="regexMatch"):
        guarded.create("HyperNode", bad_re)
    bad_tier = HyperNode(meta=ObjectMeta(name="badtier"), tier=0, members=[
        HyperNodeMember(type="Node",
                        selector=MemberSelector(exact_match=["n1"]))])
    with pytest.raises(AdmissionError, match="tier"):
        guarded.create("HyperNode", bad_tier)
