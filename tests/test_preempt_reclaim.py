"""Preempt / reclaim actions (reference actions/preempt, actions/reclaim)."""

from volcano_amd.api.types import TaskStatus
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.cache import StoreBinder
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk(actions=None):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    if actions:
        config.actions = actions
    sched = Scheduler(cache, config)
    return store, binder, cache, sched


PREEMPT = ["enqueue", "allocate", "preempt", "backfill"]
RECLAIM = ["enqueue", "allocate", "reclaim", "backfill"]


def test_priority_preemption():
    store, binder, cache, sched = mk(PREEMPT)
    for n in synth.make_nodes(2, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    # low-prio job fills the cluster
    synth.make_gang(store, "low", replicas=4, min_member=1, cpu_milli=1000,
                    mem=GI, priority=1)
    sched.run_once()
    assert len(binder.binds) == 4
    # high-prio gang arrives: needs 2 slots → preempt 2 low victims
    synth.make_gang(store, "high", replicas=2, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert len(binder.evictions) == 2
    high = cache.jobs["default/high"]
    assert high.waiting_count == 2          # pipelined onto freed nodes
    # victims' job kept at least minAvailable=1
    low = cache.jobs["default/low"]
    assert low.occupied_count >= 1


def test_preemption_respects_gang_min():
    store, binder, cache, sched = mk(PREEMPT)
    for n in synth.make_nodes(1, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    # low-prio gang with minMember == replicas: untouchable
    synth.make_gang(store, "low", replicas=2, cpu_milli=1000, mem=GI,
                    priority=1)
    sched.run_once()
    synth.make_gang(store, "high", replicas=1, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert binder.evictions == []           # would break low's gang
    assert cache.jobs["default/high"].waiting_count == 0


def test_no_preemption_same_priority():
    store, binder, cache, sched = mk(PREEMPT)
    for n in synth.make_nodes(1, cpu_milli=1000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "a", replicas=1, cpu_milli=1000, mem=GI, priority=5)
    sched.run_once()
    synth.make_gang(store, "b", replicas=1, cpu_milli=1000, mem=GI, priority=5)
    sched.run_once()
    assert binder.evictions == []


def test_gang_preempt_all_or_nothing():
    """High gang needs 3 but only 2 victims are evictable → no evictions."""
    store, binder, cache, sched = mk(PREEMPT)
    for n in synth.make_nodes(2, cpu_milli=1000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "low", replicas=2, min_member=0, cpu_milli=1000,
                    mem=GI, priority=1)
    sched.run_once()
    assert len(binder.binds) == 2
    synth.make_gang(store, "high", replicas=3, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert binder.evictions == []           # statement discarded
    assert cache.jobs["default/high"].waiting_count == 0
    assert cache.jobs["default/low"].occupied_count == 2


def test_reclaim_cross_queue():
    store, binder, cache, sched = mk(RECLAIM)
    for n in synth.make_nodes(1, cpu_milli=10000, mem=64 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("qa", weight=1))
    # qa alone: grabs the whole cluster (elastic, min 1)
    synth.make_gang(store, "ja", replicas=10, min_member=1, queue="qa",
                    cpu_milli=1000, mem=GI)
    sched.run_once()
    assert len(binder.binds) == 10
    # qb appears with equal weight → deserves half; reclaim for its gang
    store.create("Queue", synth.make_queue("qb", weight=1))
    synth.make_gang(store, "jb", replicas=3, queue="qb", cpu_milli=1000,
                    mem=GI)
    sched.run_once()
    assert len(binder.evictions) == 3
    assert cache.jobs["default/jb"].waiting_count == 3


def test_reclaim_respects_reclaimable_flag():
    store, binder, cache, sched = mk(RECLAIM)
    for n in synth.make_nodes(1, cpu_milli=4000, mem=64 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("qa", weight=1, reclaimable=False))
    synth.make_gang(store, "ja", replicas=4, min_member=1, queue="qa",
                    cpu_milli=1000, mem=GI)
    sched.run_once()
    store.create("Queue", synth.make_queue("qb", weight=1))
    synth.make_gang(store, "jb", replicas=2, queue="qb", cpu_milli=1000,
                    mem=GI)
    sched.run_once()
    assert binder.evictions == []           # qa is not reclaimable


def test_eviction_converges_to_bind():
    """Full loop with StoreBinder: evicted pods fail, preemptor binds on
    the next cycles (pipelined reservation → pending → allocated)."""
    store = ObjectStore()
    cache = SchedulerCache(store=store)     # StoreBinder
    config = default_config()
    config.actions = PREEMPT
    sched = Scheduler(cache, config)
    for n in synth.make_nodes(1, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "low", replicas=2, min_member=1, cpu_milli=1000,
                    mem=GI, priority=1)
    sched.run_once()
    synth.make_gang(store, "high", replicas=1, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    # a victim was marked Failed in the store
    failed = [p for p in store.list("Pod") if p.phase == "Failed"]
    assert len(failed) == 1
    # kubelet/GC analog: failed pod disappears
    store.delete("Pod", failed[0].meta.namespace, failed[0].meta.name)
    sched.run_once()
    high_pod = [p for p in store.list("Pod")
                if p.meta.name.startswith("high") and p.node_name]
    assert len(high_pod) == 1


def test_nominated_node_fast_path_after_preempt():
    """Preempt pipelines the gang; once victims terminate, the next cycle
    commits on the nominated nodes WITHOUT re-scoring (allocate fast
    path, reference allocate.go:797 NominatedNodeName)."""
    from volcano_amd.utils.metrics import METRICS
    store = ObjectStore()
    cache = SchedulerCache(store=store)          # StoreBinder: evict deletes
    config = default_config()
    config.actions = PREEMPT
    sched = Scheduler(cache, config)
    for n in synth.make_nodes(2, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    synth.make_gang(store, "low", replicas=4, min_member=1, cpu_milli=1000,
                    mem=GI, priority=1)
    sched.run_once()
    synth.make_gang(store, "high", replicas=2, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()                             # evict + pipeline
    high = cache.jobs["default/high"]
    assert high.waiting_count == 2
    nominated = sorted(t.node_name
                       for t in high.tasks.values() if t.node_name)
    # evicted pods: StoreBinder marked them Failed; drop them fully so
    # the capacity is free (kubelet termination analog)
    for p in list(store.list("Pod")):
        if p.phase == "Failed":
            store.delete("Pod", p.meta.namespace, p.meta.name)
    before = METRICS.counter("allocate:nominated_fastpath")
    sched.run_once()                             # fast-path commit
    assert METRICS.counter("allocate:nominated_fastpath") == before + 1
    placed = sorted(t.node_name for t in high.tasks.values())
    assert placed == nominated                   # stayed on freed nodes
    assert high.occupied_count == 2


def test_preemption_storm_converges():
    """Scaled-down storm (benchmark/storm.py at full size on GPU):
    hundreds of starving high-prio gangs preempt a saturated cluster
    within a few cycles with drift-free accounting."""
    import sys, os
    sys.path.insert(0, os.path.join(os.path.dirname(__file__), ".."))
    from benchmark.storm import run
    placed, cycles, bad = run(nodes=100, use_gpu=False, max_cycles=6)
    assert bad == 0
    assert placed == 100, f"storm placed {placed}/100 in {cycles} cycles"
    assert cycles <= 5
