"""Task-topology bucket scoring (reference plugins/task-topology
topology.go:138 calcBucketScore): affine roles pack onto the nodes of
already-placed bucket members via the per-class kernel bias plane;
anti-affine roles repel."""

from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def mk(args):
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    config.tiers[1].plugins.append(PluginOption("task-topology",
                                                arguments=args))
    sched = Scheduler(cache, config)
    for i in range(4):
        store.create("Node", synth.make_node(
            f"n{i}", cpu_milli=8000, mem=32 * GI))
    store.create("Queue", synth.make_queue("default"))
    return store, binder, cache, sched


def test_affine_worker_follows_ps():
    store, binder, cache, sched = mk({"affinity": [["ps", "worker"]]})
    # cycle 1: the ps lands somewhere (free choice)
    pg = synth.make_podgroup("tj", min_member=1)
    store.create("PodGroup", pg)
    store.create("Pod", synth.make_pod("tj-ps-0", "tj", cpu_milli=500,
                                       mem=GI, role="ps"))
    sched.run_once()
    ps_node = binder.binds["default/tj-ps-0"]
    # cycle 2: workers of the same job must co-locate with the ps even
    # though every other node is emptier (least-requested would spread)
    for i in range(2):
        store.create("Pod", synth.make_pod(f"tj-worker-{i}", "tj",
                                           cpu_milli=500, mem=GI,
                                           role="worker"))
    sched.run_once()
    assert binder.binds["default/tj-worker-0"] == ps_node
    assert binder.binds["default/tj-worker-1"] == ps_node


def test_anti_affine_replicas_repel():
    store, binder, cache, sched = mk({"anti-affinity": [["worker"]]})
    pg = synth.make_podgroup("aj", min_member=1)
    store.create("PodGroup", pg)
    store.create("Pod", synth.make_pod("aj-worker-0", "aj", cpu_milli=500,
                                       mem=GI, role="worker"))
    sched.run_once()
    first = binder.binds["default/aj-worker-0"]
    store.create("Pod", synth.make_pod("aj-worker-1", "aj", cpu_milli=500,
                                       mem=GI, role="worker"))
    sched.run_once()
    assert binder.binds["default/aj-worker-1"] != first


def test_annotation_driven_affinity():
    """Per-job groups via the reference's podgroup annotation
    (util.go:36 volcano.sh/task-topology-affinity)."""
    store, binder, cache, sched = mk({})
    pg = synth.make_podgroup("an", min_member=1)
    pg.meta.annotations["volcano.sh/task-topology-affinity"] = "ps,worker"
    store.create("PodGroup", pg)
    store.create("Pod", synth.make_pod("an-ps-0", "an", cpu_milli=500,
                                       mem=GI, role="ps"))
    sched.run_once()
    ps_node = binder.binds["default/an-ps-0"]
    store.create("Pod", synth.make_pod("an-worker-0", "an", cpu_milli=500,
                                       mem=GI, role="worker"))
    sched.run_once()
    assert binder.binds["default/an-worker-0"] == ps_node


def test_bias_does_not_leak_across_jobs():
    store, binder, cache, sched = mk({"affinity": [["ps", "worker"]]})
    pg = synth.make_podgroup("j1", min_member=1)
    store.create("PodGroup", pg)
    store.create("Pod", synth.make_pod("j1-ps-0", "j1", cpu_milli=500,
                                       mem=GI, role="ps"))
    sched.run_once()
    # an unrelated job's worker has no bucket members → spreads freely;
    # the scheduler must not pin it to j1's ps node
    pg2 = synth.make_podgroup("j2", min_member=1)
    store.create("PodGroup", pg2)
    store.create("Pod", synth.make_pod("j2-worker-0", "j2", cpu_milli=500,
                                       mem=GI, role="worker"))
    sched.run_once()
    assert binder.binds["default/j2-worker-0"] != \
        binder.binds["default/j1-ps-0"]
