"""DRA: DeviceClass/ResourceClaim scheduling through dense dims and
per-queue device-class quotas (reference
capacity.go:107-197 draQuotaAttr + predicates dynamicresources wrap)."""

from volcano_amd.api.objects import (DeviceClass, ObjectMeta, Queue,
                                     QueueSpec, ResourceClaim)
from volcano_amd.api.resource import CPU, Resource
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk(capacity_plugin=False):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    if capacity_plugin:
        # capacity replaces proportion (reference: modern quota plugin)
        for tier in config.tiers:
            tier.plugins = [p for p in tier.plugins
                            if p.name != "proportion"]
        config.tiers[1].plugins.append(PluginOption("capacity"))
    sched = Scheduler(cache, config)
    return store, binder, cache, sched


def add_gpu_nodes(store, n, gpus=8):
    for i in range(n):
        node = synth.make_node(f"n{i}", cpu_milli=32000, mem=128 * GI)
        node.meta.annotations["dra.volcano.sh/mi355x.amd.com"] = str(gpus)
        store.create("Node", node)


def mk_claim(store, name, device_class="mi355x.amd.com", count=2):
    store.create("ResourceClaim", ResourceClaim(
        meta=ObjectMeta(name=name, namespace="default"),
        device_class_name=device_class, count=count))


def test_claims_become_dense_dims_and_gate_fit():
    store, binder, cache, sched = mk()
    store.create("DeviceClass", DeviceClass(
        meta=ObjectMeta(name="mi355x.amd.com"), driver="amdgpu"))
    add_gpu_nodes(store, 2, gpus=8)
    store.create("Queue", synth.make_queue("default"))
    # gang of 5 pods x 2 devices = 10 devices; cluster offers 16 -> fits
    pg = synth.make_podgroup("dra-job", min_member=5)
    store.create("PodGroup", pg)
    for i in range(5):
        mk_claim(store, f"c-{i}")
        pod = synth.make_pod(f"dra-job-w-{i}", "dra-job", cpu_milli=1000,
                             mem=GI)
        pod.resource_claims = [f"c-{i}"]
        store.create("Pod", pod)
    sched.run_once()
    assert binder.bound_count == 5
    # per-node device capacity respected: <= 4 pods (8 devices) per node
    from collections import Counter
    per_node = Counter(binder.binds.values())
    assert all(v <= 4 for v in per_node.values()), per_node


def test_device_class_fit_blocks_when_exhausted():
    store, binder, cache, sched = mk()
    add_gpu_nodes(store, 1, gpus=4)
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("big", min_member=3)
    store.create("PodGroup", pg)
    for i in range(3):
        mk_claim(store, f"b-{i}", count=2)   # 6 devices > 4 available
        pod = synth.make_pod(f"big-w-{i}", "big", cpu_milli=500, mem=GI)
        pod.resource_claims = [f"b-{i}"]
        store.create("Pod", pod)
    sched.run_once()
    assert binder.bound_count == 0          # gang atomic: nothing placed


def test_queue_device_class_quota_capacity():
    """Fractional-GPU gangs under per-queue device quotas: queue ml is
    capped at count/mi355x.amd.com=4 even though nodes offer 16."""
    store, binder, cache, sched = mk(capacity_plugin=True)
    add_gpu_nodes(store, 2, gpus=8)
    store.create("Queue", Queue(
        meta=ObjectMeta(name="ml"),
        spec=QueueSpec(weight=1, capability=Resource(
            {CPU: 1e9, "count/mi355x.amd.com": 4.0}))))
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("q1", queue="ml", min_member=2)
    store.create("PodGroup", pg)
    for i in range(2):
        mk_claim(store, f"q1-{i}", count=2)
        pod = synth.make_pod(f"q1-w-{i}", "q1", queue="ml", cpu_milli=500,
                             mem=GI)
        pod.resource_claims = [f"q1-{i}"]
        store.create("Pod", pod)
    sched.run_once()
    assert binder.bound_count == 2          # 4 devices = exactly the quota
    # second gang in the same queue: quota exhausted -> nothing binds
    pg2 = synth.make_podgroup("q2", queue="ml", min_member=2)
    store.create("PodGroup", pg2)
    for i in range(2):
        mk_claim(store, f"q2-{i}", count=2)
        pod = synth.make_pod(f"q2-w-{i}", "q2", queue="ml", cpu_milli=500,
                             mem=GI)
        pod.resource_claims = [f"q2-{i}"]
        store.create("Pod", pod)
    sched.run_once()
    assert binder.bound_count == 2, dict(binder.binds)


def test_vgpu_fractional_gang_under_quota():
    """Fractional vGPU slices (memory+cores) pack per card while the
    queue's device-class quota caps the whole-device dimension."""
    store, binder, cache, sched = mk()
    node = synth.make_node("g0", cpu_milli=32000, mem=128 * GI)
    node.meta.annotations["volcano.sh/gpu-count"] = "2"
    node.meta.annotations["volcano.sh/gpu-memory-per-card"] = "294912"
    store.create("Node", node)
    store.create("Queue", synth.make_queue("default"))
    config = default_config()
    config.tiers[1].plugins.append(PluginOption("deviceshare"))
    sched = Scheduler(cache, config)
    pg = synth.make_podgroup("vg", min_member=3)
    store.create("PodGroup", pg)
    for i in range(3):
        pod = synth.make_pod(f"vg-w-{i}", "vg", cpu_milli=500, mem=GI)
        pod.meta.annotations["volcano.sh/vgpu-number"] = "1"
        pod.meta.annotations["volcano.sh/vgpu-memory"] = "147456"  # half card
        pod.meta.annotations["volcano.sh/vgpu-cores"] = "50"
        store.create("Pod", pod)
    sched.run_once()
    assert binder.bound_count == 3
    # card indices assigned; no card over 100% cores (2 slices max)
    from collections import Counter
    assigned = Counter()
    for p in store.list("Pod"):
        idx = p.meta.annotations.get("volcano.sh/gpu-index")
        if idx:
            for c in idx.split(","):
                assigned[c] += 1
    assert sum(assigned.values()) == 3
    assert all(v <= 2 for v in assigned.values()), assigned
