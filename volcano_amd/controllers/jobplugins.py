"""In-pod job plugins (reference ``pkg/controllers/job/plugins/``: env,
svc, ssh, distributed-framework/{tensorflow,pytorch,mpi,ray}).

The reference mutates pod specs (env vars, volumes, hosts ConfigMaps).
Pods here are scheduling-shaped dataclasses without containers, so the
plugins express the same contracts through annotations — the information
a launcher inside the pod would read (rank, master address, host list) is
present and testable.
"""

from __future__ import annotations

from typing import Callable, Dict, List


def env_plugin(job, task_spec, pod, index: int, args: List[str]) -> None:
    """reference job/plugins/env: VC_TASK_INDEX / VK_TASK_INDEX."""
    pod.meta.annotations["env/VC_TASK_INDEX"] = str(index)
    pod.meta.annotations["env/VC_JOB_NAME"] = job.meta.name


def svc_plugin(job, task_spec, pod, index: int, args: List[str]) -> None:
    """reference job/plugins/svc: headless service + hosts file — here the
    sibling host list in canonical subdomain form."""
    hosts = []
    for ts in job.spec.tasks:
        for i in range(ts.replicas):
            hosts.append(f"{job.meta.name}-{ts.name}-{i}.{job.meta.name}")
    pod.meta.annotations["svc/hosts"] = ",".join(hosts)
    pod.meta.annotations["svc/hostname"] = \
        f"{job.meta.name}-{task_spec.name}-{index}.{job.meta.name}"


def ssh_plugin(job, task_spec, pod, index: int, args: List[str]) -> None:
    """reference job/plugins/ssh: shared keypair secret volume."""
    pod.meta.annotations["ssh/secret"] = f"{job.meta.name}-ssh"


def pytorch_plugin(job, task_spec, pod, index: int, args: List[str]) -> None:
    """reference distributed-framework/pytorch: MASTER_ADDR/RANK/WORLD_SIZE."""
    master_task = args[0] if args else "master"
    world = sum(t.replicas for t in job.spec.tasks)
    rank = 0
    for ts in job.spec.tasks:
        if ts.name == task_spec.name:
            rank += index
            break
        rank += ts.replicas
    pod.meta.annotations["env/MASTER_ADDR"] = \
        f"{job.meta.name}-{master_task}-0.{job.meta.name}"
    pod.meta.annotations["env/RANK"] = str(rank)
    pod.meta.annotations["env/WORLD_SIZE"] = str(world)


def tensorflow_plugin(job, task_spec, pod, index: int, args: List[str]) -> None:
    """reference distributed-framework/tensorflow: TF_CONFIG."""
    import json
    cluster = {}
    for ts in job.spec.tasks:
        cluster[ts.name] = [
            f"{job.meta.name}-{ts.name}-{i}.{job.meta.name}:2222"
            for i in range(ts.replicas)]
    tf_config = {"cluster": cluster,
                 "task": {"type": task_spec.name, "index": index}}
    pod.meta.annotations["env/TF_CONFIG"] = json.dumps(tf_config)


def mpi_plugin(job, task_spec, pod, index: int, args: List[str]) -> None:
    """reference distributed-framework/mpi: hostfile for the master."""
    workers = []
    for ts in job.spec.tasks:
        if ts.name != task_spec.name or True:
            workers.extend(
                f"{job.meta.name}-{ts.name}-{i}.{job.meta.name}"
                for i in range(ts.replicas))
    pod.meta.annotations["mpi/hostfile"] = "\n".join(workers)


JOB_PLUGINS: Dict[str, Callable] = {
    "env": env_plugin,
    "svc": svc_plugin,
    "ssh": ssh_plugin,
    "pytorch": pytorch_plugin,
    "tensorflow": tensorflow_plugin,
    "mpi": mpi_plugin,
}
