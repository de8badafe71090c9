"""Deviceshare (GPU pools), numaaware, extender plugins."""

import json
import threading

from volcano_amd.api.devices import (ANN_ASSIGNED, DeviceRequest,
                                     GPUDevicePool)
from volcano_amd.api.objects import Numatopology, NumaZone, ObjectMeta
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk(*plugins):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    for name, args in plugins:
        config.tiers[1].plugins.append(PluginOption(name, arguments=args))
    sched = Scheduler(cache, config)
    return store, binder, cache, sched


def test_device_pool_packing():
    node = synth.make_node("g", extra={})
    node.meta.annotations["volcano.sh/gpu-count"] = "2"
    node.meta.annotations["volcano.sh/gpu-memory-per-card"] = "1000"
    pool = GPUDevicePool(node)
    # two 600-MiB slices cannot share one 1000-MiB card
    req = DeviceRequest(count=2, memory=600)
    picked = pool.allocate(req)
    assert picked is not None and len(set(picked)) == 2
    # a third 600 slice no longer fits anywhere
    assert pool.fit(DeviceRequest(count=1, memory=600)) is None
    # but a 400 slice does
    assert pool.fit(DeviceRequest(count=1, memory=400)) is not None
    pool.release(req, picked)
    assert pool.fit(DeviceRequest(count=1, memory=600)) is not None


def test_deviceshare_scheduling():
    store, binder, cache, sched = mk(("deviceshare", {}))
    plain = synth.make_node("plain", cpu_milli=8000, mem=32 * GI)
    gpu = synth.make_node("gpu", cpu_milli=8000, mem=32 * GI)
    gpu.meta.annotations["volcano.sh/gpu-count"] = "1"
    gpu.meta.annotations["volcano.sh/gpu-memory-per-card"] = "1000"
    store.create("Node", plain)
    store.create("Node", gpu)
    store.create("Queue", synth.make_queue("default"))

    pg = synth.make_podgroup("dj", min_member=1)
    store.create("PodGroup", pg)
    pod = synth.make_pod("dj-w-0", "dj", cpu_milli=500, mem=GI)
    pod.meta.annotations["volcano.sh/gpu-number"] = "1"
    pod.meta.annotations["volcano.sh/gpu-memory"] = "700"
    store.create("Pod", pod)
    sched.run_once()
    assert binder.binds["default/dj-w-0"] == "gpu"
    # card index was assigned
    task = cache.jobs["default/dj"].tasks["default/dj-w-0"]
    assert task.pod.meta.annotations[ANN_ASSIGNED] == "0"

    # second 700-MiB slice can't fit the same card → unschedulable
    pg2 = synth.make_podgroup("dj2", min_member=1)
    store.create("PodGroup", pg2)
    pod2 = synth.make_pod("dj2-w-0", "dj2", cpu_milli=500, mem=GI)
    pod2.meta.annotations["volcano.sh/gpu-number"] = "1"
    pod2.meta.annotations["volcano.sh/gpu-memory"] = "700"
    store.create("Pod", pod2)
    sched.run_once()
    assert "default/dj2-w-0" not in binder.binds


def test_numa_single_numa_node():
    store, binder, cache, sched = mk(("numaaware", {}))
    # node with 2 NUMA zones of 4 cpu each (8 total)
    n = synth.make_node("numa", cpu_milli=8000, mem=32 * GI)
    store.create("Node", n)
    store.create("Numatopology", Numatopology(
        meta=ObjectMeta(name="numa"),
        zones=[NumaZone(0, 4000, 16 * GI), NumaZone(1, 4000, 16 * GI)]))
    store.create("Queue", synth.make_queue("default"))

    # 6-cpu single-numa pod cannot fit any one zone despite 8 free
    pg = synth.make_podgroup("nj", min_member=1)
    store.create("PodGroup", pg)
    pod = synth.make_pod("nj-w-0", "nj", cpu_milli=6000, mem=GI)
    pod.meta.annotations["volcano.sh/numa-topology-policy"] = "single-numa-node"
    store.create("Pod", pod)
    sched.run_once()
    assert "default/nj-w-0" not in binder.binds

    # a 3-cpu one fits and gets pinned
    pg2 = synth.make_podgroup("nj2", min_member=1)
    store.create("PodGroup", pg2)
    pod2 = synth.make_pod("nj2-w-0", "nj2", cpu_milli=3000, mem=GI)
    pod2.meta.annotations["volcano.sh/numa-topology-policy"] = "single-numa-node"
    store.create("Pod", pod2)
    sched.run_once()
    assert binder.binds["default/nj2-w-0"] == "numa"
    task = cache.jobs["default/nj2"].tasks["default/nj2-w-0"]
    assert task.pod.meta.annotations["volcano.sh/numa-node"] in ("0", "1")


def test_extender_predicate_http():
    # tiny sidecar: only node "allowed" passes the predicate
    from http.server import BaseHTTPRequestHandler, HTTPServer

    class Handler(BaseHTTPRequestHandler):
        def do_POST(self):
            n = int(self.headers.get("Content-Length", 0))
            self.rfile.read(n)
            body = b"{}"
            if self.path.endswith("/predicate"):
                body = json.dumps({"nodes": ["allowed"]}).encode()
            self.send_response(200)
            self.send_header("Content-Type", "application/json")
            self.send_header("Content-Length", str(len(body)))
            self.end_headers()
            self.wfile.write(body)

        def log_message(self, *a):
            pass

    srv = HTTPServer(("127.0.0.1", 0), Handler)
    port = srv.server_address[1]
    t = threading.Thread(target=srv.serve_forever, daemon=True)
    t.start()
    try:
        store, binder, cache, sched = mk(("extender", {
            "extender.urlPrefix": f"http://127.0.0.1:{port}",
            "extender.predicateVerb": "predicate"}))
        store.create("Node", synth.make_node("allowed", cpu_milli=8000,
                                             mem=32 * GI))
        store.create("Node", synth.make_node("denied", cpu_milli=8000,
                                             mem=32 * GI))
        store.create("Queue", synth.make_queue("default"))
        synth.make_gang(store, "ej", replicas=2, cpu_milli=1000, mem=GI)
        sched.run_once()
        assert len(binder.binds) == 2
        assert set(binder.binds.values()) == {"allowed"}
    finally:
        srv.shutdown()


def test_interpod_anti_affinity_excludes_conode():
    store, binder, cache, sched = mk(("interpodaffinity", {}))
    for n in synth.make_nodes(3, cpu_milli=8000, mem=32 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    pg = synth.make_podgroup("aa", min_member=3)
    store.create("PodGroup", pg)
    for i in range(3):
        pod = synth.make_pod(f"aa-w-{i}", "aa", cpu_milli=500, mem=GI)
        pod.affinity = {"podAntiAffinity": {"group": "replica-set"}}
        store.create("Pod", pod)
    sched.run_once()
    assert len(binder.binds) == 3
    # each member on its own node
    assert len(set(binder.binds.values())) == 3
    # a 4th member has no node left
    pg2 = synth.make_podgroup("aa2", min_member=1)
    store.create("PodGroup", pg2)
    pod = synth.make_pod("aa2-w-0", "aa2", cpu_milli=500, mem=GI)
    pod.affinity = {"podAntiAffinity": {"group": "replica-set"}}
    store.create("Pod", pod)
    sched.run_once()
    assert "default/aa2-w-0" not in binder.binds


def test_interpod_affinity_colocates():
    store, binder, cache, sched = mk(("interpodaffinity", {}))
    for n in synth.make_nodes(3, cpu_milli=8000, mem=32 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    # anchor pod of group "svc"
    pg = synth.make_podgroup("anchor", min_member=1)
    store.create("PodGroup", pg)
    anchor = synth.make_pod("anchor-w-0", "anchor", cpu_milli=500, mem=GI)
    anchor.affinity = {"podAffinity": {"group": "svc"}}
    store.create("Pod", anchor)
    sched.run_once()
    anchor_node = binder.binds["default/anchor-w-0"]
    # follower must land on the anchor's node
    pg2 = synth.make_podgroup("fol", min_member=1)
    store.create("PodGroup", pg2)
    fol = synth.make_pod("fol-w-0", "fol", cpu_milli=500, mem=GI)
    fol.affinity = {"podAffinity": {"group": "svc"}}
    store.create("Pod", fol)
    sched.run_once()
    assert binder.binds["default/fol-w-0"] == anchor_node


def test_dynamic_bits_not_stale_across_cycles():
    """Regression: per-session node sets (device-unfit etc.) must not
    linger once the condition clears (pod deletes don't repack planes)."""
    store, binder, cache, sched = mk(("deviceshare", {}))
    gpu = synth.make_node("gpu", cpu_milli=8000, mem=32 * GI)
    gpu.meta.annotations["volcano.sh/gpu-count"] = "1"
    gpu.meta.annotations["volcano.sh/gpu-memory-per-card"] = "1000"
    store.create("Node", gpu)
    store.create("Queue", synth.make_queue("default"))

    def dev_pod(name, pg):
        p = synth.make_pod(name, pg, cpu_milli=500, mem=GI)
        p.meta.annotations["volcano.sh/gpu-number"] = "1"
        p.meta.annotations["volcano.sh/gpu-memory"] = "700"
        return p

    pg = synth.make_podgroup("d1", min_member=1)
    store.create("PodGroup", pg)
    store.create("Pod", dev_pod("d1-w-0", "d1"))
    sched.run_once()
    assert binder.binds["default/d1-w-0"] == "gpu"

    pg2 = synth.make_podgroup("d2", min_member=1)
    store.create("PodGroup", pg2)
    store.create("Pod", dev_pod("d2-w-0", "d2"))
    sched.run_once()
    assert "default/d2-w-0" not in binder.binds     # card full

    store.delete("Pod", "default", "d1-w-0")        # releases the slices
    sched.run_once()
    assert binder.binds.get("default/d2-w-0") == "gpu"


def test_numa_cpuset_provider_assignment():
    """cpuset providers (reference numaaware/provider, VERDICT r1 plugin
    depth): pinned pods get concrete CPU ids from their zone's pool,
    exclusive across pods; the node agent enforces them via
    cgroup cpuset.cpus."""
    from volcano_amd.api.objects import Numatopology, NumaZone, ObjectMeta
    from volcano_amd.scheduler import (FakeBinder, Scheduler,
                                       SchedulerCache, default_config)
    from volcano_amd.scheduler.config import PluginOption
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    from volcano_amd.utils.features import set_gates

    GI = 1024 ** 3
    set_gates({"ResourceTopology": True})
    try:
        store = ObjectStore()
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder)
        config = default_config()
        config.tiers[1].plugins.append(PluginOption("numaaware"))
        sched = Scheduler(cache, config)
        store.create("Node", synth.make_node("n1", cpu_milli=16000,
                                             mem=64 * GI))
        store.create("Numatopology", Numatopology(
            meta=ObjectMeta(name="n1"),
            zones=[NumaZone(id=0, cpu_milli=8000, cpus=list(range(8))),
                   NumaZone(id=1, cpu_milli=8000,
                            cpus=list(range(8, 16)))]))
        store.create("Queue", synth.make_queue("default"))
        pg = synth.make_podgroup("numa", min_member=2)
        store.create("PodGroup", pg)
        for i in range(2):
            pod = synth.make_pod(f"numa-w-{i}", "numa", cpu_milli=4000,
                                 mem=GI)
            pod.meta.annotations["volcano.sh/numa-topology-policy"] = \
                "single-numa-node"
            store.create("Pod", pod)
        sched.run_once()
        assert binder.bound_count == 2
        seen = []
        for p in store.list("Pod"):
            cs = p.meta.annotations.get("volcano.sh/cpuset")
            if cs:
                ids = [int(x) for x in cs.split(",")]
                assert len(ids) == 4            # ceil(4000m) = 4 cores
                zone = int(p.meta.annotations["volcano.sh/numa-node"])
                lo, hi = (0, 8) if zone == 0 else (8, 16)
                assert all(lo <= c < hi for c in ids)
                seen.extend(ids)
        assert len(seen) == 8
        assert len(set(seen)) == 8              # exclusive assignment
    finally:
        set_gates({"ResourceTopology": False})
