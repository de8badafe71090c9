"""Gangpreempt/gangreclaim bundle eviction + resourcequota plugin."""

from volcano_amd.api.objects import ObjectMeta, ResourceQuota
from volcano_amd.api.resource import CPU, MEMORY, Resource
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk(actions=None, extra_plugins=()):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    if actions:
        config.actions = actions
    for name in extra_plugins:
        config.tiers[1].plugins.append(PluginOption(name))
    sched = Scheduler(cache, config)
    return store, binder, cache, sched


def test_gangpreempt_evicts_whole_bundle():
    store, binder, cache, sched = mk(
        actions=["enqueue", "allocate", "gangpreempt", "backfill"])
    for n in synth.make_nodes(2, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    # low-prio gang with minMember == replicas fills the cluster —
    # task-level preemption could never touch it (gang min); BUNDLE
    # preemption evicts the whole job
    synth.make_gang(store, "low", replicas=4, cpu_milli=1000, mem=GI,
                    priority=1)
    sched.run_once()
    assert len(binder.binds) == 4
    synth.make_gang(store, "high", replicas=3, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert len(binder.evictions) == 4      # entire low gang evicted
    assert cache.jobs["default/high"].waiting_count == 3


def test_gangpreempt_noop_when_insufficient():
    store, binder, cache, sched = mk(
        actions=["enqueue", "allocate", "gangpreempt", "backfill"])
    for n in synth.make_nodes(1, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "low", replicas=2, cpu_milli=1000, mem=GI,
                    priority=1)
    sched.run_once()
    # preemptor needs 4 cpu; cluster only has 2 → no point evicting
    synth.make_gang(store, "high", replicas=4, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert binder.evictions == []
    assert cache.jobs["default/low"].occupied_count == 2


def test_resourcequota_blocks_enqueue():
    store, binder, cache, sched = mk(extra_plugins=("resourcequota",))
    for n in synth.make_nodes(2, cpu_milli=16000, mem=64 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    store.create("ResourceQuota", ResourceQuota(
        meta=ObjectMeta(name="ns-quota"),
        hard=Resource({CPU: 3000.0, MEMORY: 64.0 * GI})))
    # 2-cpu job fits the quota
    synth.make_gang(store, "ok", replicas=2, cpu_milli=1000, mem=GI)
    # 4-cpu job exceeds it
    synth.make_gang(store, "toobig", replicas=4, cpu_milli=1000, mem=GI)
    sched.run_once()
    assert sum(1 for k in binder.binds if k.startswith("default/ok")) == 2
    assert not any(k.startswith("default/toobig") for k in binder.binds)
    assert cache.jobs["default/toobig"].phase == "Pending"


def test_gangpreempt_honors_pdb_veto():
    """plugin vetoes are authoritative for bundles —
    a PDB covering the low-prio gang's pods blocks gangpreempt even
    though the victims are strictly lower priority."""
    from volcano_amd.api.objects import ObjectMeta, PodDisruptionBudget
    store, binder, cache, sched = mk(
        actions=["enqueue", "allocate", "gangpreempt", "backfill"],
        extra_plugins=("pdb",))
    for n in synth.make_nodes(2, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    store.create("PodDisruptionBudget", PodDisruptionBudget(
        meta=ObjectMeta(name="protect-low", namespace="default"),
        selector={"app": "low"}, min_available=4))
    pg = synth.make_gang(store, "low", replicas=4, cpu_milli=1000, mem=GI,
                         priority=1)
    for p in store.list("Pod"):
        if p.meta.name.startswith("low"):
            p.meta.labels["app"] = "low"
            store.update("Pod", p)
    sched.run_once()
    # mark bound pods running so the PDB sees them healthy
    for p in store.list("Pod"):
        if p.meta.name.startswith("low"):
            p.phase = "Running"
            store.update("Pod", p)
    sched.run_once()
    synth.make_gang(store, "high", replicas=3, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert binder.evictions == []      # PDB veto holds for the bundle


def test_gangpreempt_requires_lower_job_priority():
    """Reference gangpreempt.go:210 — equal-priority jobs are not bundle
    victims (the old OR fallback would still have evicted via the
    plugin intersection)."""
    store, binder, cache, sched = mk(
        actions=["enqueue", "allocate", "gangpreempt", "backfill"])
    for n in synth.make_nodes(2, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "low", replicas=4, cpu_milli=1000, mem=GI,
                    priority=50)
    sched.run_once()
    synth.make_gang(store, "high", replicas=3, cpu_milli=1000, mem=GI,
                    priority=50)
    sched.run_once()
    assert binder.evictions == []


def test_evict_task_credits_releasing():
    """cache.evict_task must flip used→releasing so
    future_idle sees the evicted capacity."""
    from volcano_amd.api.info import NodeInfo, JobInfo, TaskInfo
    from volcano_amd.api.resource import CPU
    from volcano_amd.api.types import TaskStatus
    cache = SchedulerCache(store=None, binder=FakeBinder())
    node = NodeInfo(synth.make_node("n1", cpu_milli=4000, mem=8 * GI))
    cache.add_node_info(node)
    pg = synth.make_podgroup("j1")
    job = JobInfo("default/j1", pg)
    pod = synth.make_pod("j1-w-0", "j1", cpu_milli=1000, mem=GI,
                         node_name="n1", phase="Running")
    t = TaskInfo.from_pod(pod, job.key)
    job.add_task(t)
    cache.add_job_info(job)
    used_before = node.used.get(CPU)
    assert node.releasing.get(CPU) == 0
    cache.evict_task(t, "test")
    assert t.status == TaskStatus.RELEASING
    assert node.releasing.get(CPU) == 1000      # credited
    assert node.used.get(CPU) == used_before    # still held until released
    fi = node.future_idle
    assert fi.get(CPU) == 4000 - used_before + 1000


def test_gangpreempt_safe_bundle_keeps_victim_gang_alive():
    """Reference bundle.go BundleSafe: the victim gang has surplus above
    minAvailable — gangpreempt evicts ONLY the surplus and the victim
    stays a live gang."""
    from volcano_amd.utils import synth
    from volcano_amd.utils.uthelper import TestCommonStruct
    G = 10 ** 9

    def pg(name, mm, prio, phase="Inqueue"):
        g = synth.make_podgroup(name, queue="q1", namespace="c1",
                                min_member=mm)
        g.status.phase = phase
        g.meta.annotations["priority"] = str(prio)
        return g

    pods = [synth.make_pod(f"low-{i}", "lowpg", namespace="c1",
                           cpu_milli=1000, mem=G, node_name="n1",
                           phase="Running", priority=1)
            for i in range(4)]
    pods.append(synth.make_pod("high-0", "highpg", namespace="c1",
                               cpu_milli=2000, mem=2 * G, priority=100))
    t = TestCommonStruct(
        name="safe bundle evicts surplus only",
        podgroups=[pg("lowpg", 2, 1, phase="Running"),
                   pg("highpg", 1, 100)],
        pods=pods,
        nodes=[synth.make_node("n1", cpu_milli=4000, mem=4 * G, pods=10)],
        queues=[synth.make_queue("q1")],
        actions=["enqueue", "gangpreempt"],
        expect_evicted=["c1/low-0", "c1/low-1"],
    ).run()
    t.check_all()
    low = t.cache.jobs["c1/lowpg"]
    assert low.occupied_count >= low.min_available   # gang survived


def test_gangpreempt_allow_whole_bundle_false_blocks_whole_job():
    """Reference gangpreempt.go:253: with allowWholeBundle=false a victim
    at its minAvailable (whole-bundle only) is never evicted."""
    from volcano_amd.utils import synth
    from volcano_amd.utils.uthelper import TestCommonStruct
    G = 10 ** 9

    def scenario(allow):
        g1 = synth.make_podgroup("lowpg", queue="q1", namespace="c1",
                                 min_member=2)
        g1.status.phase = "Running"
        g1.meta.annotations["priority"] = "1"
        g2 = synth.make_podgroup("highpg", queue="q1", namespace="c1",
                                 min_member=1)
        g2.status.phase = "Inqueue"
        g2.meta.annotations["priority"] = "100"
        pods = [synth.make_pod(f"low-{i}", "lowpg", namespace="c1",
                               cpu_milli=2000, mem=2 * G, node_name="n1",
                               phase="Running", priority=1)
                for i in range(2)]       # at min: no surplus
        pods.append(synth.make_pod("high-0", "highpg", namespace="c1",
                                   cpu_milli=3000, mem=3 * G, priority=100))
        t = TestCommonStruct(
            podgroups=[g1, g2], pods=pods,
            nodes=[synth.make_node("n1", cpu_milli=4000, mem=4 * G,
                                   pods=10)],
            queues=[synth.make_queue("q1")],
            actions=["enqueue", "gangpreempt"],
            cycles=0)           # build only — configure, then cycle
        t.run()
        return t

    t = scenario(False)
    t.scheduler.config.configurations = {
        "gangpreempt": {"allowWholeBundle": False}}
    t.scheduler.run_once()
    assert t.binder.evictions == [], t.binder.evictions

    t2 = scenario(True)     # default allowWholeBundle=true evicts both
    t2.scheduler.run_once()
    assert sorted(t2.binder.evictions) == ["c1/low-0", "c1/low-1"], \
        t2.binder.evictions


def test_gangpreempt_parse_arguments_defaults():
    """Reference gangpreempt_test.go TestParseArguments +
    InvalidMaxDomainsFallsBackToDefault."""
    from volcano_amd.scheduler.actions.gangpreempt import (
        DEFAULT_MAX_DOMAINS, GangPreemptAction)

    class FakeSsn:
        class config:
            configurations = {"gangpreempt": {"maxDomains": 3,
                                              "allowWholeBundle": False}}

    a = GangPreemptAction()
    a._parse_arguments(FakeSsn)
    assert a.max_domains == 3 and a.allow_whole_bundle is False

    for bad in (0, -1, "x", None):
        FakeSsn.config.configurations = {"gangpreempt": {"maxDomains": bad}}
        a._parse_arguments(FakeSsn)
        assert a.max_domains == DEFAULT_MAX_DOMAINS
        assert a.allow_whole_bundle is True
