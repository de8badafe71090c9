"""Edge-case coverage: cron field semantics, store journal trimming,
watch kind filtering, resource vector edges, bit-plane overflow."""

import time

import pytest

from volcano_amd.api.resource import Resource, ResourceDims
from volcano_amd.scheduler.tensors import BitRegistry, NodeTensors
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.cron import CronSchedule


def test_cron_fields():
    # minute steps
    s = CronSchedule("*/15 * * * *")
    t = time.mktime((2026, 9, 12, 10, 30, 0, 0, 0, -1))
    assert s.matches(t)
    assert not s.matches(t + 60)
    # ranges + lists
    s = CronSchedule("0 9-17 * * 1-5")
    mon_10am = time.mktime((2026, 9, 14, 10, 0, 0, 0, 0, -1))   # Monday
    sun_10am = time.mktime((2026, 9, 13, 10, 0, 0, 0, 0, -1))   # Sunday
    assert s.matches(mon_10am)
    assert not s.matches(sun_10am)
    # next_after lands on a matching minute
    nxt = s.next_after(sun_10am)
    assert s.matches(nxt)
    with pytest.raises(ValueError):
        CronSchedule("* * *")


def test_store_journal_trim_and_filter():
    s = ObjectStore(journal_size=10)
    for i in range(25):
        s.create("Node", synth.make_node(f"n-{i:03d}"))
    evs = s.journal_since(0)
    assert len(evs) == 10                       # ring-trimmed
    assert evs[-1][0] == s.resource_version
    # kind filter
    s.create("Queue", synth.make_queue("q"))
    assert all(k == "Node" for _, _, k, _ in s.journal_since(0, ("Node",)))


def test_bit_registry_overflow():
    reg = BitRegistry(max_words=1)
    for i in range(64):
        reg.bit(f"k{i}")
    with pytest.raises(OverflowError):
        reg.bit("one-too-many")


def test_resource_from_vector_roundtrip_zero():
    dims = ResourceDims()
    r = Resource({})
    vec = r.to_vector(dims)
    assert Resource.from_vector(vec, dims).is_empty()


def test_node_tensors_dynamic_bit_width_growth():
    from volcano_amd.api.info import NodeInfo
    dims = ResourceDims()
    nt = NodeTensors(dims, label_words=2)
    nodes = [NodeInfo(n) for n in synth.make_nodes(4)]
    nt.pack(nodes)
    # burn through word 0 to force a second plane word
    for i in range(70):
        nt.add_dynamic_bit(f"grow-{i}", [0])
    assert nt.planes_t.shape[0] >= 2
    assert nt.planes_np.shape == tuple(nt.planes_t.shape)
    # require vector padding matches
    words = nt.bit_words([0, 69])
    assert len(words) == nt.labels.words


def test_agent_scheduler_skips_gang_pods():
    from volcano_amd.scheduler.agentscheduler import AgentScheduler
    store = ObjectStore()
    store.create("Node", synth.make_node("n1"))
    asched = AgentScheduler(store, workers=1)
    # a gang pod (has podgroup annotation) is NOT fast-path eligible
    store.create("Pod", synth.make_pod("gangpod", podgroup="pg1"))
    assert asched.run_once() == 0


def test_hip_plan_rejects_too_many_dims():
    """The HIP path's revert/commit kernels map one 64-lane wave per log
    entry; >64 resource dims must fail loudly, not silently skip dims."""
    import numpy as np
    import pytest
    import torch
    from volcano_amd.api.resource import ResourceDims
    from volcano_amd.scheduler.plan import CyclePlan, ClassPlan, run_plan_hip
    from volcano_amd.scheduler.tensors import NodeTensors
    from volcano_amd.api.info import NodeInfo, TaskClass
    from volcano_amd.utils import synth

    dims = ResourceDims()
    for i in range(70):
        dims.add(f"paa:g{i}")
    nt = NodeTensors(dims)
    nt.pack([NodeInfo(synth.make_node("n0"))])
    assert nt.r > 64
    plan = CyclePlan(nt, torch.full((1, nt.r), 1e18), torch.zeros(1, nt.r))
    tc = TaskClass(signature="s", role="", request=None, tasks=[object()])
    plan.classes.append(ClassPlan(
        tclass=tc, job_key="j", queue_idx=0,
        req=np.zeros(nt.r, dtype=np.float32), tolerated=-1,
        require=np.zeros(1, dtype=np.int64),
        forbid=np.zeros(1, dtype=np.int64), min_needed=0))
    with pytest.raises(RuntimeError, match="64 resource dims"):
        run_plan_hip(plan)


def test_many_synthetic_dims_cpu_cycle():
    """A cycle with >16 dims (anti-affinity + host ports) through the
    torch oracle — the dim count that overflowed the GPU kernel's
    per-thread array before the fix."""
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    GI = 1024 ** 3
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    sched = Scheduler(cache, default_config())
    for i in range(6):
        store.create("Node", synth.make_node(f"n{i}", cpu_milli=8000,
                                             mem=32 * GI))
    store.create("Queue", synth.make_queue("default"))
    for j in range(12):
        synth.make_gang(store, f"g{j}", replicas=1, cpu_milli=500, mem=GI)
        pod = store.get("Pod", "default", f"g{j}-worker-0")
        pod.affinity = {"podAntiAffinity": {"group": f"grp{j}"}}
        pod.host_ports = [8000 + j]
        store.update("Pod", pod)
    sched.run_once()
    assert cache.dims and len(cache.dims.names) > 16
    assert len(binder.binds) == 12


def test_sorted_jobs_lexsort_equals_tuple_sort():
    """The columnar (lexsort) job ordering must equal the tuple sort
    exactly, including creation-time and key tie-breaks."""
    import random
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    rng = random.Random(4)
    store = ObjectStore()
    cache = SchedulerCache(store=store, binder=FakeBinder())
    sched = Scheduler(cache, default_config())
    store.create("Node", synth.make_node("n0"))
    store.create("Queue", synth.make_queue("default"))
    for j in range(400):      # >= 256 triggers the columnar path
        synth.make_gang(store, f"ls{j:04d}", replicas=1, cpu_milli=10,
                        priority=rng.randint(0, 3))
    ssn = sched.open_session()
    jobs = list(ssn.jobs.values())
    rng.shuffle(jobs)
    fast = [j.key for j in ssn.sorted_jobs(jobs)]
    keys = [k for tier in ssn.job_order_keys for k in tier]
    slow = [j.key for j in sorted(
        jobs, key=lambda j: tuple(k(j) for k in keys)
        + (j.creation_timestamp, j.key))]
    assert fast == slow
    sched.close_session(ssn)


def test_scheduler_determinism():
    """Two scheduler instances over identical inventories make identical
    decisions — the decision plane has no hidden randomness (ties break
    structurally: lower node index, stable job order)."""
    from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                       default_config)
    from volcano_amd.store import ObjectStore
    from volcano_amd.utils import synth
    import random

    def run():
        rng = random.Random(77)
        store = ObjectStore()
        binder = FakeBinder()
        cache = SchedulerCache(store=store, binder=binder)
        sched = Scheduler(cache, default_config())
        for i in range(20):
            store.create("Node", synth.make_node(f"n{i:02d}",
                                                 cpu_milli=8000))
        store.create("Queue", synth.make_queue("qa", weight=2))
        store.create("Queue", synth.make_queue("qb", weight=1))
        for j in range(40):
            synth.make_gang(store, f"d{j:03d}",
                            replicas=rng.randint(1, 4),
                            queue=rng.choice(["qa", "qb"]),
                            cpu_milli=rng.choice([500, 1000]),
                            priority=rng.randint(0, 3))
        sched.run_once()
        sched.run_once()
        return dict(binder.binds)

    assert run() == run()
