"""Concurrency smoke: writers, controller loop and scheduler loop share
the store from different threads (SURVEY §5 race-detection analog — the
design is single-writer-per-cycle with mutex-guarded store/cache; this
exercises the locking)."""

import random
import threading
import time

from volcano_amd.api.objects import Job, JobSpec, ObjectMeta, TaskSpec
from volcano_amd.api.resource import CPU
from volcano_amd.controllers import ControllerManager
from volcano_amd.scheduler import Scheduler, SchedulerCache
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.kubelet import FakeKubelet

GI = 1024 ** 3


def test_threaded_store_scheduler_controllers():
    store = ObjectStore()
    for n in synth.make_nodes(50, cpu_milli=16000, mem=64 * GI):
        store.create("Node", n)
    cm = ControllerManager(store, ["job", "podgroup", "queue",
                                   "garbagecollector"], period=0.01)
    cache = SchedulerCache(store=store)
    sched = Scheduler(cache)
    kubelet = FakeKubelet(store)
    errors = []
    stop = threading.Event()

    def writer(tid):
        rng = random.Random(tid)
        i = 0
        try:
            while not stop.is_set():
                i += 1
                store.create("Job", Job(
                    meta=ObjectMeta(name=f"w{tid}-{i:04d}"),
                    spec=JobSpec(ttl_seconds_after_finished=0.0,
                                 tasks=[TaskSpec(
                                     name="w",
                                     replicas=rng.randint(1, 3),
                                     template={"resources": {
                                         "cpu": "1", "memory": "1Gi"}})])))
                time.sleep(0.002)
        except Exception as e:           # pragma: no cover
            errors.append(("writer", e))

    def kubelet_loop():
        try:
            while not stop.is_set():
                kubelet.tick(complete=lambda p: "Succeeded"
                             if random.random() < 0.3 else None)
                time.sleep(0.005)
        except Exception as e:           # pragma: no cover
            errors.append(("kubelet", e))

    cm.run()
    threads = [threading.Thread(target=writer, args=(t,)) for t in range(3)]
    threads.append(threading.Thread(target=kubelet_loop))
    for t in threads:
        t.start()
    try:
        deadline = time.time() + 2.0
        cycles = 0
        while time.time() < deadline:
            sched.run_once()
            cycles += 1
    finally:
        stop.set()
        for t in threads:
            t.join(timeout=5)
        cm.stop()

    assert not errors, errors
    assert cycles >= 3
    # settle and check invariants
    cm.sync_until_quiet(200)
    sched.run_once()
    for ni in cache.nodes.values():
        rec = sum(t.request.get(CPU) for t in ni.tasks.values()
                  if t.status.occupies_node)
        assert abs(rec - ni.used.get(CPU)) < 1.0
        assert ni.used.get(CPU) <= ni.allocatable.get(CPU) + 1.0
