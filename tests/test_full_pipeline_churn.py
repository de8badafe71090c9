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
    config.tiers[1].plugins.append(PluginOption(
        "task-topology", arguments={"affinity": [["ps", "worker"]]}))
    sched = Scheduler(cache, config)

    from volcano_amd.api.objects import (ObjectMeta, PersistentVolume,
                                         PersistentVolumeClaim, ZONE_LABEL)
    for i in range(12):
        store.create("Node", synth.make_node(
            f"n-{i:03d}", cpu_milli=8000, mem=32 * GI,
            labels={ZONE_LABEL: f"z{i % 3}"}))
    store.create("PersistentVolume", PersistentVolume(
        meta=ObjectMeta(name="pv-z1", labels={ZONE_LABEL: "z1"})))
    store.create("PersistentVolumeClaim", PersistentVolumeClaim(
        meta=ObjectMeta(name="claim-z1", namespace="default"),
        volume_name="pv-z1"))
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
            roll = rng.random()
            synth.make_gang(store, f"c{jid:04d}",
                            replicas=rng.randint(1, 6),
                            min_member=None if rng.random() < 0.5 else 1,
                            queue=rng.choice(["qa", "qb"]),
                            cpu_milli=rng.choice([500, 1000, 2000]),
                            mem=GI, priority=rng.randint(0, 20),
                            role=rng.choice(["ps", "worker", "etl"]))
            if roll < 0.2:
                # sprinkle the new predicate surfaces into the churn
                for p in store.list("Pod"):
                    if p.meta.name.startswith(f"c{jid:04d}-"):
                        if roll < 0.07:
                            p.volumes = ["claim-z1"]
                        elif roll < 0.14:
                            p.affinity = {"exists": [ZONE_LABEL]}
                        else:
                            p.affinity = {"gt": {"missing-numeric": 5}}
                        store.update("Pod", p)
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
            # the incremental counter must equal a from-scratch recount
            from volcano_amd.api.info import _OCC_SET
            recount = sum(1 for t in job.tasks.values()
                          if t.status in _OCC_SET)
            assert occ == recount, \
                f"cycle {cycle}: {job.key} occ counter drift {occ} vs {recount}"
            # gang atomicity (pipelined reservations count toward min)
            assert occ == 0 or occ + job.waiting_count >= \
                min(job.min_available, len(job.tasks)), \
                f"cycle {cycle}: {job.key} partial {occ}/{job.min_available}"

    assert len(binder.binds) > 50       # the pipeline kept flowing
