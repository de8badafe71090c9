"""Feature gates (reference pkg/features/volcano_features.go): optional
behaviors flip off cleanly via conf YAML / env, defaults all-on."""

import pytest

from volcano_amd.utils import features


@pytest.fixture(autouse=True)
def _reset_gates():
    yield
    features.reset()


def test_defaults_on_and_unknown_on():
    assert features.enabled("VolcanoJobSupport")
    assert features.enabled("SomeFutureGate")    # forward-compatible


def test_set_and_reset():
    features.set_gates({"PriorityClass": False})
    assert not features.enabled("PriorityClass")
    features.reset()
    assert features.enabled("PriorityClass")


def test_yaml_plumbing():
    from volcano_amd.scheduler.config import SchedulerConfiguration
    conf = SchedulerConfiguration.from_yaml(
        "actions: \"enqueue, allocate\"\n"
        "feature_gates:\n  CSIStorage: false\n  PriorityClass: true\n")
    assert conf.feature_gates == {"CSIStorage": False, "PriorityClass": True}


def test_workload_gate_stops_pod_wrapping():
    from volcano_amd.controllers import ControllerManager
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    features.set_gates({"WorkLoadSupport": False})
    store = ObjectStore()
    cm = ControllerManager(store, ["podgroup"])
    store.create("Pod", synth.make_pod("solo", podgroup=""))
    cm.sync_until_quiet()
    assert store.list("PodGroup") == []          # no implicit group
    features.set_gates({"WorkLoadSupport": True})
    store.create("Pod", synth.make_pod("solo2", podgroup=""))
    cm.sync_until_quiet()
    assert len(store.list("PodGroup")) == 1


def test_csistorage_gate_disables_volume_zone():
    from volcano_amd.api.objects import (ObjectMeta, PersistentVolume,
                                         PersistentVolumeClaim, ZONE_LABEL)
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    GI = 1024 ** 3

    def run(csi_on):
        features.set_gates({"CSIStorage": csi_on})
        store = ObjectStore()
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder)
        sched = Scheduler(cache, default_config())
        store.create("Node", synth.make_node(
            "a", cpu_milli=8000, mem=32 * GI, labels={ZONE_LABEL: "z1"}))
        store.create("Queue", synth.make_queue("default"))
        store.create("PersistentVolume", PersistentVolume(
            meta=ObjectMeta(name="pv", labels={ZONE_LABEL: "z9"}),
            capacity=GI))
        store.create("PersistentVolumeClaim", PersistentVolumeClaim(
            meta=ObjectMeta(name="c", namespace="default"),
            volume_name="pv"))
        synth.make_gang(store, "vj", replicas=1, cpu_milli=500, mem=GI)
        pod = store.get("Pod", "default", "vj-worker-0")
        pod.volumes = ["c"]
        store.update("Pod", pod)
        sched.run_once()
        return "default/vj-worker-0" in binder.binds

    assert not run(True)     # zone mismatch blocks when the filter is on
    assert run(False)        # gate off: zone ignored, pod places


def test_gate_via_scheduler_config():
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    config = default_config()
    config.feature_gates = {"PriorityClass": False}
    Scheduler(SchedulerCache(store=ObjectStore(), binder=FakeBinder()),
              config)
    assert not features.enabled("PriorityClass")


def test_env_parsing(monkeypatch):
    monkeypatch.setenv("VAMD_FEATURE_GATES",
                       "CSIStorage=false, PriorityClass=true,Bad")
    features.reset()        # re-reads the env
    assert not features.enabled("CSIStorage")
    assert features.enabled("PriorityClass")
    assert features.enabled("WorkLoadSupport")   # untouched default
    monkeypatch.delenv("VAMD_FEATURE_GATES")
    features.reset()
    assert features.enabled("CSIStorage")
