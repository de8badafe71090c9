"""Kubernetes wire-surface parity: group/version API
paths, k8s-shaped manifests, CRD manifests, streaming watch."""

import json

import yaml
from fastapi.testclient import TestClient

from volcano_amd.store import ObjectStore
from volcano_amd.store.apiserver import create_app
from volcano_amd.store.k8s import (GVK, api_version, crd_manifest,
                                   from_manifest, to_manifest)


def mk_client():
    store = ObjectStore()
    app = create_app(store)
    return store, TestClient(app)


def test_examples_job_applies_through_group_version_path():
    store, client = mk_client()
    from volcano_amd.utils import synth
    store.create("Queue", synth.make_queue("default"))
    with open("examples/job.k8s.yaml") as f:
        manifest = yaml.safe_load(f)
    r = client.post("/apis/batch.volcano.sh/v1alpha1/namespaces/default/jobs",
                    json=manifest)
    assert r.status_code == 200, r.text
    out = r.json()
    assert out["apiVersion"] == "batch.volcano.sh/v1alpha1"
    assert out["kind"] == "Job"
    assert out["metadata"]["name"] == "tf-training"
    # stored object has the internal shape with fields translated
    job = store.get("Job", "default", "tf-training")
    assert job.spec.min_available == 3
    assert job.spec.max_retry == 3
    assert [t.name for t in job.spec.tasks] == ["ps", "worker"]
    assert job.spec.tasks[1].min_available == 2
    # GET round-trips the k8s shape
    r = client.get("/apis/batch.volcano.sh/v1alpha1/namespaces/default/"
                   "jobs/tf-training")
    assert r.status_code == 200
    got = r.json()
    assert got["spec"]["minAvailable"] == 3
    assert got["spec"]["tasks"][0]["template"]["resources"]["cpu"] == "2"
    # list shape
    r = client.get("/apis/batch.volcano.sh/v1alpha1/namespaces/default/jobs")
    assert r.json()["kind"] == "JobList"
    assert len(r.json()["items"]) == 1
    # delete
    r = client.delete("/apis/batch.volcano.sh/v1alpha1/namespaces/default/"
                      "jobs/tf-training")
    assert r.status_code == 200
    assert store.get("Job", "default", "tf-training") is None


def test_queue_and_core_paths():
    store, client = mk_client()
    r = client.post("/apis/scheduling.volcano.sh/v1beta1/queues", json={
        "apiVersion": "scheduling.volcano.sh/v1beta1", "kind": "Queue",
        "metadata": {"name": "ml"},
        "spec": {"weight": 4, "reclaimable": False,
                 "capability": {"cpu": 100000.0}},
    })
    assert r.status_code == 200, r.text
    q = store.get("Queue", "default", "ml")
    assert q.spec.weight == 4 and q.spec.reclaimable is False
    # core group: node via /api/v1
    r = client.post("/api/v1/nodes", json={
        "apiVersion": "v1", "kind": "Node", "metadata": {"name": "n1"},
        "allocatable": {"cpu": 8000.0, "memory": 1.0e9},
    })
    assert r.status_code == 200, r.text
    r = client.get("/api/v1/nodes")
    assert r.json()["kind"] == "NodeList"
    assert len(r.json()["items"]) == 1


def test_admission_applies_on_k8s_path():
    store, client = mk_client()
    bad = {"apiVersion": "batch.volcano.sh/v1alpha1", "kind": "Job",
           "metadata": {"name": "bad", "namespace": "default"},
           "spec": {"minAvailable": -1, "tasks": []}}
    r = client.post("/apis/batch.volcano.sh/v1alpha1/namespaces/default/jobs",
                    json=bad)
    assert r.status_code == 400
    assert "admission denied" in r.text


def test_streaming_watch():
    store, client = mk_client()
    store.create("Queue", from_manifest({
        "apiVersion": "scheduling.volcano.sh/v1beta1", "kind": "Queue",
        "metadata": {"name": "w1"}, "spec": {"weight": 1}}))
    with client.stream(
            "GET", "/apis/scheduling.volcano.sh/v1beta1/queues",
            params={"watch": 1, "since": 0}) as r:
        line = next(r.iter_lines())
        ev = json.loads(line)
    assert ev["type"] == "ADDED"
    assert ev["object"]["kind"] == "Queue"
    assert ev["object"]["metadata"]["name"] == "w1"


def test_crd_manifests_cover_all_groups():
    import os
    for kind, (g, v, plural) in GVK.items():
        man = crd_manifest(kind)
        if not g or g in ("policy", "resource.k8s.io"):
            assert man is None
            continue
        assert man["metadata"]["name"] == f"{plural}.{g}"
        assert man["spec"]["names"]["kind"] == kind
        ver = man["spec"]["versions"][0]
        assert ver["name"] == v and ver["served"] and ver["storage"]
        assert ver["schema"]["openAPIV3Schema"]["type"] == "object"
        # the generated file is committed in deploy/crds/
        path = os.path.join("deploy", "crds", f"{plural}.{g}.yaml")
        assert os.path.exists(path), f"missing {path}"
        with open(path) as f:
            assert yaml.safe_load(f) == man


def test_manifest_round_trip_preserves_opaque_maps():
    m = {"apiVersion": api_version("PodGroup"), "kind": "PodGroup",
         "metadata": {"name": "pg", "namespace": "default",
                      "labels": {"volcano.sh/job-name": "x"}},
         "spec": {"minMember": 2, "queue": "ml",
                  "minResources": {"cpu": 2000.0},
                  "networkTopology": {"mode": "hard",
                                      "highestTierAllowed": 1}}}
    obj = from_manifest(m)
    assert obj.spec.min_member == 2
    assert obj.spec.network_topology == {"mode": "hard",
                                         "highestTierAllowed": 1}
    assert obj.meta.labels == {"volcano.sh/job-name": "x"}
    back = to_manifest(obj)
    assert back["spec"]["minMember"] == 2
    assert back["metadata"]["labels"] == {"volcano.sh/job-name": "x"}
    assert back["spec"]["networkTopology"]["highestTierAllowed"] == 1


def test_datadependency_group_wire_paths():
    """datadependency/v1alpha1 over the k8s wire: create a cluster-scoped
    DataSource and a namespaced claim, list both back, and let the
    controller bind them."""
    store, client = mk_client()
    r = client.post(
        "/apis/datadependency.volcano.sh/v1alpha1/datasources", json={
            "apiVersion": "datadependency.volcano.sh/v1alpha1",
            "kind": "DataSource",
            "metadata": {"name": "sales", "namespace": ""},
            "system": "hive", "type": "table", "name": "db.sales",
            "clusterNames": ["c1"],
        })
    assert r.status_code == 200, r.text
    r = client.post(
        "/apis/datadependency.volcano.sh/v1alpha1/namespaces/ml/"
        "datasourceclaims", json={
            "apiVersion": "datadependency.volcano.sh/v1alpha1",
            "kind": "DataSourceClaim",
            "metadata": {"name": "train-in", "namespace": "ml"},
            "system": "hive", "dataSourceType": "table",
            "dataSourceName": "db.sales",
            "workload": {"apiVersion": "batch.volcano.sh/v1alpha1",
                         "kind": "Job", "name": "train"},
        })
    assert r.status_code == 200, r.text
    r = client.get(
        "/apis/datadependency.volcano.sh/v1alpha1/datasources")
    assert len(r.json()["items"]) == 1

    from volcano_amd.controllers import ControllerManager
    mgr = ControllerManager(store, controllers=["datadependency"])
    mgr.sync_once()
    c = store.get("DataSourceClaim", "ml", "train-in")
    assert c.phase == "Bound" and c.bound_data_source == "sales"
