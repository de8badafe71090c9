#!/usr/bin/env python3
"""Stability soak: continuous churn cycles (controllers + scheduler +
kubelet) for --seconds, asserting node-accounting invariants and flat
device memory at the end.  Run on a GPU box for the HIP path."""

import argparse
import random
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

from volcano_amd.api.objects import Job, JobSpec, ObjectMeta, TaskSpec
from volcano_amd.api.resource import CPU
from volcano_amd.controllers import ControllerManager
from volcano_amd.scheduler import Scheduler, SchedulerCache, default_config
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.kubelet import FakeKubelet

GI = 1024 ** 3


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--seconds", type=float, default=60.0)
    ap.add_argument("--nodes", type=int, default=5000)
    ap.add_argument("--jobs-per-cycle", type=int, default=20)
    args = ap.parse_args()

    rng = random.Random(7)
    use_gpu = torch.cuda.is_available()
    store = ObjectStore()
    for n in synth.make_nodes(args.nodes, cpu_milli=16000, mem=64 * GI):
        store.create("Node", n)
    cm = ControllerManager(store, ["job", "podgroup", "queue",
                                   "garbagecollector"])
    config = default_config()
    config.use_hip = use_gpu
    config.device = "cuda" if use_gpu else "cpu"
    config.actions = ["enqueue", "allocate", "preempt", "backfill"]
    cache = SchedulerCache(store=store, device=config.device)
    sched = Scheduler(cache, config)
    kubelet = FakeKubelet(store)

    def mk_job(i):
        return Job(meta=ObjectMeta(name=f"s-{i:05d}"),
                   spec=JobSpec(
                       ttl_seconds_after_finished=0.0,
                       tasks=[TaskSpec(
                           name="w", replicas=rng.randint(1, 8),
                           template={"resources": {
                               "cpu": str(rng.choice([1, 2, 4])),
                               "memory": "2Gi"}})]))

    t0 = time.time()
    i = cycles = 0
    mem0 = None
    while time.time() - t0 < args.seconds:
        for _ in range(args.jobs_per_cycle):
            i += 1
            store.create("Job", mk_job(i))
        cm.sync_until_quiet()
        sched.run_once()
        kubelet.tick()
        running = [j for j in store.list("Job")
                   if j.status.phase == "Running"]
        for victim in running[: len(running) // 2]:
            nm = victim.meta.name
            kubelet.tick(complete=lambda p, n=nm: "Succeeded"
                         if p.meta.labels.get("volcano.sh/job-name") == n
                         else None)
        cm.sync_until_quiet()
        cycles += 1
        if cycles == 5 and use_gpu:
            mem0 = torch.cuda.memory_allocated()
    if use_gpu:
        torch.cuda.synchronize()
    mem1 = torch.cuda.memory_allocated() if use_gpu else 0

    bad = 0
    for ni in cache.nodes.values():
        rec = sum(t.request.get(CPU) for t in ni.tasks.values()
                  if t.status.occupies_node)
        if abs(rec - ni.used.get(CPU)) > 1.0 or \
                ni.used.get(CPU) > ni.allocatable.get(CPU) + 1.0:
            bad += 1
    print(f"cycles={cycles} jobs={i} left={store.count('Job')} "
          f"bad_nodes={bad} mem_growth={mem1 - (mem0 or mem1)}")
    assert bad == 0, "node accounting drift"
    if mem0 is not None:
        assert mem1 - mem0 < 50 * 1024 * 1024, "device memory growth"
    print("SOAK OK")
    return 0


if __name__ == "__main__":
    sys.exit(main())
