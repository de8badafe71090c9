#!/usr/bin/env python3
"""Preemption storm: the cluster is saturated with
low-priority gangs, then THOUSANDS of higher-priority gangs arrive at
once.  Measures cycles-to-quiescence, evictions and accounting drift —
the regime the reference's bundle/ROI machinery exists for
(actions/utils/bundle.go:232-303)."""

import argparse
import sys
import time

sys.path.insert(0, __file__.rsplit("/", 2)[0])

import torch

from volcano_amd.api.resource import CPU
from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def run(nodes=5000, low_jobs=None, storm_jobs=None, use_gpu=None,
        max_cycles=8):
    # one 2-pod low gang per node (minMember=1): every gang holds exactly
    # one pod ABOVE its gang minimum — the storm's preemptable capacity
    # is `nodes` slots, so storm_jobs = nodes converges iff preempt's
    # victim selection works at scale under gang-min protection
    low_jobs = nodes if low_jobs is None else low_jobs
    storm_jobs = nodes if storm_jobs is None else storm_jobs
    use_gpu = torch.cuda.is_available() if use_gpu is None else use_gpu
    device = "cuda" if use_gpu else "cpu"
    store = ObjectStore()
    binder = FakeBinder()
    cache = SchedulerCache(store=store, binder=binder, device=device)
    config = default_config()
    config.use_hip = use_gpu
    config.device = device
    config.actions = ["enqueue", "allocate", "preempt", "backfill"]
    sched = Scheduler(cache, config)

    for n in synth.make_nodes(nodes, cpu_milli=8000, mem=32 * GI):
        store.create("Node", n)
    store.create("Queue", synth.make_queue("default"))
    # low-prio fill: 2 pods x 4 cpu per job -> saturates the cluster
    for j in range(low_jobs):
        synth.make_gang(store, f"low-{j:05d}", replicas=2, min_member=1,
                        cpu_milli=4000, mem=GI, priority=1)
    sched.run_once()
    filled = binder.bound_count
    # storm: every high-prio gang needs one 4-cpu slot
    for j in range(storm_jobs):
        synth.make_gang(store, f"hi-{j:05d}", replicas=1, cpu_milli=4000,
                        mem=GI, priority=100)
    t0 = time.perf_counter()
    cycles = 0
    placed = 0
    while cycles < max_cycles:
        cycles += 1
        sched.run_once()
        # evicted victims terminate between cycles (kubelet analog)
        for key in binder.evictions:
            ns, name = key.split("/")
            if store.get("Pod", ns, name) is not None:
                store.delete("Pod", ns, name)
        binder.evictions.clear()
        placed = sum(1 for k in binder.binds if k.startswith("default/hi-"))
        if placed >= storm_jobs:
            break
    elapsed = time.perf_counter() - t0

    # accounting drift check
    bad = 0
    for ni in cache.nodes.values():
        rec = sum(t.request.get(CPU) for t in ni.tasks.values()
                  if t.status.occupies_node)
        if abs(rec - ni.used.get(CPU)) > 1.0 or \
                ni.used.get(CPU) > ni.allocatable.get(CPU) + 1.0:
            bad += 1
    print(f"storm: filled={filled} placed={placed}/{storm_jobs} "
          f"cycles={cycles} elapsed={elapsed:.2f}s bad_nodes={bad}")
    return placed, cycles, bad


if __name__ == "__main__":
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=5000)
    ap.add_argument("--low-jobs", type=int, default=None)
    ap.add_argument("--storm-jobs", type=int, default=None)
    ap.add_argument("--cpu", action="store_true")
    args = ap.parse_args()
    placed, cycles, bad = run(args.nodes, args.low_jobs, args.storm_jobs,
                              use_gpu=(False if args.cpu else None))
    want = args.storm_jobs if args.storm_jobs is not None else args.nodes
    assert bad == 0, "accounting drift"
    assert placed == want, "storm did not converge"
    print("STORM OK")
