"""GPU full-stack churn: controllers + HIP decision plane + kubelet on a
moderate synthetic cluster, multi-cycle, with invariants — the
whole-framework-on-hardware test the round-end driver replays."""

import random

import pytest
import torch

pytestmark = pytest.mark.gpu

GI = 1024 ** 3


def test_gpu_full_stack_churn():
    from volcano_amd.api.resource import CPU
    from volcano_amd.api.types import JobPhase
    from volcano_amd.controllers import ControllerManager
    from volcano_amd.ops import hip as vamd_hip
    from volcano_amd.scheduler import (Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    from volcano_amd.utils.kubelet import FakeKubelet
    from tests.test_controllers import mk_job

    vamd_hip._load()      # HIP library must be present — no silent fallback
    rng = random.Random(1)
    store = ObjectStore()
    for n in synth.make_nodes(500, cpu_milli=16000, mem=64 * GI):
        store.create("Node", n)
    cm = ControllerManager(store, ["job", "podgroup", "queue",
                                   "garbagecollector"])
    config = default_config()
    config.use_hip = True
    config.device = "cuda"
    cache = SchedulerCache(store=store, device="cuda")
    sched = Scheduler(cache, config)
    kubelet = FakeKubelet(store)

    submitted = 0
    for cycle in range(12):
        for _ in range(10):          # 10 new gangs per cycle
            submitted += 1
            store.create("Job", mk_job(
                f"g-{submitted:04d}", replicas=rng.randint(1, 6),
                cpu=str(rng.choice([1, 2, 4])),
                ttl_seconds_after_finished=0.0))
        cm.sync_until_quiet()
        sched.run_once()
        kubelet.tick()
        if cycle % 2 == 1:
            running = [j for j in store.list("Job")
                       if j.status.phase == JobPhase.RUNNING.value]
            for victim in running[: len(running) // 2]:
                name = victim.meta.name
                kubelet.tick(complete=lambda p, n=name:
                             "Succeeded" if p.meta.labels.get(
                                 "volcano.sh/job-name") == n else None)
        cm.sync_until_quiet()

    torch.cuda.synchronize()
    # invariants after 12 churn cycles on the HIP path
    bound = sum(1 for p in store.list("Pod") if p.node_name)
    assert bound > 0
    for ni in cache.nodes.values():
        recomputed = sum(t.request.get(CPU) for t in ni.tasks.values()
                         if t.status.occupies_node)
        assert abs(recomputed - ni.used.get(CPU)) < 1.0
        assert ni.used.get(CPU) <= ni.allocatable.get(CPU) + 1.0
    # device tensors agree with the host mirror
    cache.ensure_packed()
    nt = cache.node_tensors
    cpu_idx = nt.dims.index[CPU]
    dev_used = nt.used_t[cpu_idx].cpu().numpy()
    for ni in sorted(cache.nodes.values(), key=lambda n: n.name):
        assert abs(dev_used[ni.node_id] - ni.used.get(CPU)) < 1.0
    # work flowed through: most submitted gangs completed and were GC'd
    assert store.count("Job") < submitted


def test_gpu_priority_preemption():
    """Preempt path on device tensors: fused candidate scan + eviction."""
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth

    store = ObjectStore()
    binder = FakeBinder()
    for n in synth.make_nodes(4, cpu_milli=2000, mem=8 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    config = default_config()
    config.use_hip = True
    config.device = "cuda"
    config.actions = ["enqueue", "allocate", "preempt", "backfill"]
    cache = SchedulerCache(store=store, binder=binder, device="cuda")
    sched = Scheduler(cache, config)

    synth.make_gang(store, "low", replicas=8, min_member=1, cpu_milli=1000,
                    mem=GI, priority=1)
    sched.run_once()
    assert len(binder.binds) == 8
    synth.make_gang(store, "high", replicas=3, cpu_milli=1000, mem=GI,
                    priority=100)
    sched.run_once()
    assert len(binder.evictions) == 3
    assert cache.jobs["default/high"].waiting_count == 3
