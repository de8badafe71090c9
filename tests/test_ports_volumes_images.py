"""nodeports / nodevolumelimits / imagelocality — the k8s filter/score
plugins the reference wraps via plugins/predicates + plugins/nodeorder,
mapped onto synthetic resource dims and kernel bias planes."""

from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def world(n_nodes=3, plugins=(), node_kw=None):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    for name, args in plugins:
        config.tiers[1].plugins.append(PluginOption(name, arguments=args))
    sched = Scheduler(cache, config)
    for i in range(n_nodes):
        store.create("Node", synth.make_node(
            f"n{i}", cpu_milli=8000, mem=32 * GI, **(node_kw or {})))
    store.create("Queue", synth.make_queue("default"))
    return store, binder, cache, sched


def mk_pod(store, name, **kw):
    extra_attrs = {k: kw.pop(k) for k in
                   ("host_ports", "image", "volumes") if k in kw}
    synth.make_gang(store, name, replicas=1, cpu_milli=500, mem=GI, **kw)
    pod = store.get("Pod", "default", f"{name}-worker-0")
    for k, v in extra_attrs.items():
        setattr(pod, k, v)
    store.update("Pod", pod)
    return f"default/{name}-worker-0"


def test_host_port_conflict_spreads():
    store, binder, cache, sched = world(n_nodes=2)
    a = mk_pod(store, "pa", host_ports=[8080])
    b = mk_pod(store, "pb", host_ports=[8080])
    sched.run_once()
    assert binder.binds[a] != binder.binds[b]     # same port → distinct nodes


def test_host_port_exhaustion_blocks():
    store, binder, cache, sched = world(n_nodes=2)
    keys = [mk_pod(store, f"p{i}", host_ports=[9000]) for i in range(3)]
    sched.run_once()
    bound = [k for k in keys if k in binder.binds]
    assert len(bound) == 2                        # only 2 nodes → 2 pods
    assert len({binder.binds[k] for k in bound}) == 2


def test_different_ports_share_a_node():
    store, binder, cache, sched = world(n_nodes=1)
    a = mk_pod(store, "qa", host_ports=[80])
    b = mk_pod(store, "qb", host_ports=[81])
    sched.run_once()
    assert binder.binds[a] == binder.binds[b] == "n0"


def test_volume_attach_limit():
    from volcano_amd.api.objects import ObjectMeta, PersistentVolumeClaim
    store, binder, cache, sched = world(n_nodes=1)
    node = store.get("Node", "default", "n0")
    node.meta.annotations["volcano.sh/max-volumes"] = "2"
    store.update("Node", node)
    for v in ("v1", "v2", "v3"):
        store.create("PersistentVolumeClaim", PersistentVolumeClaim(
            meta=ObjectMeta(name=v, namespace="default")))
    a = mk_pod(store, "va", volumes=["v1", "v2"])
    b = mk_pod(store, "vb", volumes=["v3"])
    sched.run_once()
    # only one of them fits under the 2-attach budget
    assert (a in binder.binds) != (b in binder.binds) or \
        (a in binder.binds and b not in binder.binds)
    n_bound = sum(1 for k in (a, b) if k in binder.binds)
    assert n_bound == 1


def test_image_locality_prefers_cached_node():
    store, binder, cache, sched = world(
        n_nodes=4, plugins=(("imagelocality", {}),))
    node = store.get("Node", "default", "n2")
    node.images = ["registry/app:v7"]
    store.update("Node", node)
    key = mk_pod(store, "img", image="registry/app:v7")
    sched.run_once()
    assert binder.binds[key] == "n2"


def test_image_locality_no_cache_no_pin():
    store, binder, cache, sched = world(
        n_nodes=3, plugins=(("imagelocality", {}),))
    key = mk_pod(store, "img2", image="registry/other:v1")
    sched.run_once()
    assert key in binder.binds                    # schedules anywhere


def test_volume_binding_assume_on_bind():
    """Unbound PVC binds to a fitting free PV in the landing node's zone
    when the pod schedules (k8s volumebinding WaitForFirstConsumer)."""
    from volcano_amd.api.objects import (ObjectMeta, PersistentVolume,
                                         PersistentVolumeClaim, ZONE_LABEL)
    store = ObjectStore()
    cache = SchedulerCache(store=store)               # StoreBinder
    sched = Scheduler(cache, default_config())
    store.create("Node", synth.make_node(
        "z1n", cpu_milli=8000, mem=32 * GI, labels={ZONE_LABEL: "z1"}))
    store.create("Queue", synth.make_queue("default"))
    # two free PVs: wrong zone (big) and right zone (fits)
    store.create("PersistentVolume", PersistentVolume(
        meta=ObjectMeta(name="pv-z2", labels={ZONE_LABEL: "z2"}),
        capacity=100 * GI))
    store.create("PersistentVolume", PersistentVolume(
        meta=ObjectMeta(name="pv-z1", labels={ZONE_LABEL: "z1"}),
        capacity=50 * GI))
    store.create("PersistentVolumeClaim", PersistentVolumeClaim(
        meta=ObjectMeta(name="want", namespace="default"),
        request=10 * GI))
    key = mk_pod(store, "wf", volumes=["want"])
    sched.run_once()
    pod = store.get("Pod", "default", "wf-worker-0")
    assert pod.node_name == "z1n"
    pvc = store.get("PersistentVolumeClaim", "default", "want")
    assert pvc.volume_name == "pv-z1"                 # zone-matched PV
    pv = store.get("PersistentVolume", "default", "pv-z1")
    assert pv.claim_ref == "default/want"
