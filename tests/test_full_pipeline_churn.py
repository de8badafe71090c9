"""Churn with the FULL action pipeline (enqueue, allocate, preempt,
reclaim, gangpreempt, backfill, shuffle) and mixed
priorities/queues/preemptable pods — interaction stress for the
corrective paths."""

import random

from volcano_amd.api.resource import CPU
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def test_full_pipeline_churn_invariants():
    rng = random.Random(3)
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder)
    config = default_config()
    config.actions = ["enqueue", "allocate", "preempt", "reclaim",
                      "gangpreempt", "backfill", "shuffle"]
    config.tiers[1].plugins.append(PluginOption("pdb"))
    config.tiers[1].plugins.append(
        PluginOption("cdp", arguments={"cdp.cooldown-time": "0s"}))
    sched = Scheduler(cache, config)

    for i in range(12):
        store.create("Node", synth.make_node(
            f"n-{i:03d}", cpu_milli=8000, mem=32 * GI))
    store.create("Queue", synth.make_queue("qa", weight=3))
    store.create("Queue", synth.make_queue("qb", weight=1))

    jid = 0
    for cycle in range(25):
        # new arrivals with varied priorities; some preemptable
        for _ in range(rng.randint(1, 4)):
            jid += 1
            kw = {}
            if rng.random() < 0.4:
                kw["phase"] = "Pending"
            synth.make_gang(store, f"c{jid:04d}",
                            replicas=rng.randint(1, 6),
                            min_member=None if rng.random() < 0.5 else 1,
                            queue=rng.choice(["qa", "qb"]),
                            cpu_milli=rng.choice([500, 1000, 2000]),
                            mem=GI, priority=rng.randint(0, 20))
        # random completions: delete a random bound pod (simulates finish)
        bound_pods = [p for p in store.list("Pod") if p.node_name]
        for p in rng.sample(bound_pods, min(3, len(bound_pods))):
            store.delete("Pod", p.meta.namespace, p.meta.name)
        sched.run_once()

        # invariants every cycle
        from volcano_amd.api.types import TaskStatus
        for ni in cache.nodes.values():
            # used covers occupying AND releasing tasks (evicted pods hold
            # their resources until they actually terminate)
            rec = sum(t.request.get(CPU) for t in ni.tasks.values()
                      if t.status.occupies_node
                      or t.status == TaskStatus.RELEASING)
            assert abs(rec - ni.used.get(CPU)) < 1.0, \
                f"cycle {cycle}: node {ni.name} drift {rec} vs {ni.used.get(CPU)}"
            assert ni.used.get(CPU) <= ni.allocatable.get(CPU) + 1.0
        for job in cache.jobs.values():
            occ = job.occupied_count
            # gang atomicity (pipelined reservations count toward min)
            assert occ == 0 or occ + job.waiting_count >= \
                min(job.min_available, len(job.tasks)), \
                f"cycle {cycle}: {job.key} partial {occ}/{job.min_available}"

    assert len(binder.binds) > 50       # the pipeline kept flowing
