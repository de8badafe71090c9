#!/usr/bin/env python3
"""End-to-end demo: controllers + scheduler + fake kubelet on a tiny
synthetic cluster — the `example/job.yaml on kind` analog
(BASELINE config #1), CPU or GPU.

Run: python examples/demo.py
"""

import os
import sys

sys.path.insert(0, os.path.dirname(os.path.dirname(os.path.abspath(__file__))))

import torch

from volcano_amd.api.objects import Job, JobSpec, ObjectMeta, TaskSpec
from volcano_amd.controllers import ControllerManager
from volcano_amd.scheduler import Scheduler, SchedulerCache, default_config
from volcano_amd.store import ObjectStore
from volcano_amd.utils import synth
from volcano_amd.utils.kubelet import FakeKubelet
from volcano_amd.webhooks import default_chain


def main():
    store = ObjectStore()
    guarded = default_chain(store).guard(store)   # admission on writes

    # a 4-node cluster
    for node in synth.make_nodes(4, cpu_milli=16000, mem=64 * 1024 ** 3):
        store.create("Node", node)

    cm = ControllerManager(store, ["job", "podgroup", "queue",
                                   "garbagecollector"])
    use_gpu = torch.cuda.is_available()
    config = default_config()
    config.use_hip = use_gpu
    config.device = "cuda" if use_gpu else "cpu"
    cache = SchedulerCache(store=store, device=config.device)
    sched = Scheduler(cache, config)
    kubelet = FakeKubelet(store)

    # a distributed-training-shaped vcjob (validated + defaulted by the
    # admission chain, expanded by the job controller)
    guarded.create("Job", Job(
        meta=ObjectMeta(name="train"),
        spec=JobSpec(
            plugins={"env": [], "svc": [], "pytorch": ["master"]},
            tasks=[
                TaskSpec(name="master", replicas=1,
                         template={"resources": {"cpu": "2",
                                                 "memory": "4Gi"}}),
                TaskSpec(name="worker", replicas=6,
                         template={"resources": {"cpu": "4",
                                                 "memory": "8Gi"}}),
            ])))

    cm.sync_until_quiet()
    print(f"controller created {store.count('Pod')} pods + podgroup "
          f"{store.get('PodGroup', 'default', 'train').spec.min_member}"
          f"-min gang")

    sched.run_once()
    bound = [p for p in store.list("Pod") if p.node_name]
    print(f"scheduler ({'HIP/gfx950' if use_gpu else 'cpu-oracle'}) bound "
          f"{len(bound)}/7 pods:")
    for p in sorted(bound, key=lambda p: p.meta.name):
        rank = p.meta.annotations.get("env/RANK", "?")
        print(f"  {p.meta.name:<18} -> {p.node_name}   RANK={rank}")

    kubelet.tick()
    cm.sync_until_quiet()
    job = store.get("Job", "default", "train")
    print(f"job phase: {job.status.phase} "
          f"(running={job.status.running})")
    assert job.status.phase == "Running"
    print("demo OK")


if __name__ == "__main__":
    main()
