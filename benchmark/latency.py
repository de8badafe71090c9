#!/usr/bin/env python3
"""Pod scheduling latency benchmark — the serving-path metric
(reference: pod_scheduling_latency_seconds from the audit exporter,
benchmark/README.md).

Drives single pods through the agent-scheduler fast path (one fused
feasibility+score pass per pod) and through the full gang cycle, and
reports p50/p99 creation→bind latency.

  python benchmark/latency.py --nodes 5000 --pods 2000
"""

from __future__ import annotations

import argparse
import os
import sys
import time

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from volcano_amd.scheduler import (FakeBinder, Scheduler, SchedulerCache,
                                   default_config)
from volcano_amd.scheduler.agentscheduler import AgentScheduler
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def pct(xs, p):
    xs = sorted(xs)
    return xs[min(len(xs) - 1, int(len(xs) * p))]


def main():
    ap = argparse.ArgumentParser()
    ap.add_argument("--nodes", type=int, default=5000)
    ap.add_argument("--pods", type=int, default=2000)
    args = ap.parse_args()
    use_gpu = torch.cuda.is_available()
    device = "cuda" if use_gpu else "cpu"

    # -- agent-scheduler fast path: pod-at-a-time ---------------------------
    store = ObjectStore()
    for n in synth.make_nodes(args.nodes, cpu_milli=32000, mem=128 * GI):
        store.create("Node", n)
    binder = FakeBinder()
    asched = AgentScheduler(store, binder=binder, workers=1, device=device)
    asched.cache.sync()
    asched.cache.ensure_packed()

    lat_fast = []
    for i in range(args.pods):
        p = synth.make_pod(f"fp-{i:05d}", podgroup="", cpu_milli=1000, mem=GI)
        p.meta.annotations.pop("scheduling.volcano.sh/group-name", None)
        t0 = time.perf_counter()
        store.create("Pod", p)
        asched.run_once()
        lat_fast.append(time.perf_counter() - t0)
    assert len(binder.binds) == args.pods, \
        f"fast path bound {len(binder.binds)}/{args.pods}"

    # -- full gang cycle: one-pod gang per cycle ----------------------------
    store2 = ObjectStore()
    for n in synth.make_nodes(args.nodes, cpu_milli=32000, mem=128 * GI):
        store2.create("Node", n)
    store2.create("Queue", synth.make_queue("default"))
    binder2 = FakeBinder()
    config = default_config()
    config.use_hip = use_gpu
    config.device = device
    cache2 = SchedulerCache(store=store2, binder=binder2, device=device)
    sched = Scheduler(cache2, config)
    sched.run_once()             # warm the session machinery

    lat_cycle = []
    n_cycle = min(args.pods, 200)
    for i in range(n_cycle):
        t0 = time.perf_counter()
        synth.make_gang(store2, f"cj-{i:05d}", replicas=1, cpu_milli=1000,
                        mem=GI)
        sched.run_once()
        lat_cycle.append(time.perf_counter() - t0)
    assert len(binder2.binds) == n_cycle

    import json
    print(json.dumps({
        "metric": "pod_scheduling_latency_seconds",
        "nodes": args.nodes,
        "device": device,
        "fast_path": {"pods": args.pods,
                      "p50_ms": round(pct(lat_fast, 0.5) * 1000, 3),
                      "p99_ms": round(pct(lat_fast, 0.99) * 1000, 3)},
        "full_cycle": {"pods": n_cycle,
                       "p50_ms": round(pct(lat_cycle, 0.5) * 1000, 3),
                       "p99_ms": round(pct(lat_cycle, 0.99) * 1000, 3)},
    }), flush=True)


if __name__ == "__main__":
    main()
