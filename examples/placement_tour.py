#!/usr/bin/env python3
"""Placement-feature tour: one tiny cluster exercising the predicate and
scoring surface end to end — zones, host ports, volume binding, image
locality, task-topology bucket packing and pod anti-affinity.

Run: python examples/placement_tour.py        (CPU or GPU)
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from volcano_amd.api.objects import (ObjectMeta, PersistentVolume,
                                     PersistentVolumeClaim, ZONE_LABEL)
from volcano_amd.scheduler import Scheduler, SchedulerCache, default_config
from volcano_amd.scheduler.config import PluginOption
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth

GI = 1024 ** 3


def main():
    device = "cuda" if torch.cuda.is_available() else "cpu"
    store = ObjectStore()
    for i, zone in enumerate(["z1", "z1", "z2", "z2"]):
        node = synth.make_node(f"n{i}", cpu_milli=16000, mem=64 * GI,
                               labels={ZONE_LABEL: zone})
        if i == 3:
            node.images = ["registry/train:v1"]
        store.create("Node", node)
    store.create("Queue", synth.make_queue("default"))
    store.create("PersistentVolume", PersistentVolume(
        meta=ObjectMeta(name="data-pv", labels={ZONE_LABEL: "z2"}),
        capacity=100 * GI))
    store.create("PersistentVolumeClaim", PersistentVolumeClaim(
        meta=ObjectMeta(name="data", namespace="default"), request=10 * GI))

    config = default_config()
    config.use_hip = device == "cuda"
    config.device = device
    config.tiers[1].plugins.append(PluginOption(
        "task-topology", arguments={"affinity": [["ps", "worker"]]}))
    config.tiers[1].plugins.append(PluginOption("imagelocality"))
    cache = SchedulerCache(store=store, device=device)
    sched = Scheduler(cache, config)

    # 1. a web gang that must spread (one host port per node)
    synth.make_gang(store, "web", replicas=3, cpu_milli=1000, mem=GI)
    for p in store.list("Pod"):
        if p.meta.name.startswith("web-"):
            p.host_ports = [443]
            store.update("Pod", p)
    # 2. a data pod whose PVC pins it to the PV's zone (z2)
    synth.make_gang(store, "etl", replicas=1, cpu_milli=1000, mem=GI)
    pod = store.get("Pod", "default", "etl-worker-0")
    pod.volumes = ["data"]
    store.update("Pod", pod)
    # 3. a trainer preferring the node with its image cached
    synth.make_gang(store, "train", replicas=1, cpu_milli=1000, mem=GI)
    pod = store.get("Pod", "default", "train-worker-0")
    pod.image = "registry/train:v1"
    store.update("Pod", pod)

    sched.run_once()

    # 4. ps lands; workers added next cycle follow it (bucket packing)
    synth.make_gang(store, "psjob", replicas=1, role="ps",
                    min_member=1, cpu_milli=500, mem=GI)
    sched.run_once()
    for w in range(2):
        store.create("Pod", synth.make_pod(
            f"psjob-worker-{w}", "psjob", role="worker",
            cpu_milli=500, mem=GI))
    sched.run_once()

    pods = {p.meta.name: p.node_name for p in store.list("Pod")}
    web_nodes = sorted(v for k, v in pods.items() if k.startswith("web-"))
    assert len(set(web_nodes)) == 3, "host-port spread failed"
    etl_zone = store.get("Node", "default", pods["etl-worker-0"]) \
        .meta.labels[ZONE_LABEL]
    assert etl_zone == "z2", "volume zone failed"
    pvc = store.get("PersistentVolumeClaim", "default", "data")
    assert pvc.volume_name == "data-pv", "volume binding failed"
    assert pods["train-worker-0"] == "n3", "image locality failed"
    ps_node = pods["psjob-ps-0"]
    assert pods["psjob-worker-0"] == ps_node == pods["psjob-worker-1"], \
        "task-topology bucket packing failed"

    print(f"placement tour OK on {device}:")
    print(f"  web (hostPort 443)  -> {web_nodes} (spread)")
    print(f"  etl (PVC->z2 PV)    -> {pods['etl-worker-0']} (zone z2, "
          f"claim bound to {pvc.volume_name})")
    print(f"  train (image cache) -> {pods['train-worker-0']}")
    print(f"  psjob bucket        -> all on {ps_node}")
    return 0


if __name__ == "__main__":
    sys.exit(main())
