"""vcctl — the CLI (reference ``pkg/cli/`` + ``cmd/cli/vcctl.go:28-45``:
job run/list/view/suspend/resume/delete, queue create/list/get/delete,
jobflow list, pod list).

Targets either a running apiserver (``--server``) or a state file
(``--state``, the etcd-snapshot analog) so it works without any daemon.
Run: ``python -m volcano_amd.cli.vcctl job list``.
"""

from __future__ import annotations

import argparse
import sys
import yaml

from ..api.objects import (Command, Job, JobSpec, ObjectMeta, Queue,
                           QueueSpec, TaskSpec, from_dict, to_dict)
from ..api.resource import Resource
from ..store import ObjectStore


class StateBackend:
    """File-backed store access (loads, mutates, saves)."""

    def __init__(self, path: str):
        self.path = path
        try:
            self.store = ObjectStore.load(path)
        except FileNotFoundError:
            self.store = ObjectStore()

    def __getattr__(self, name):
        return getattr(self.store, name)

    def flush(self):
        self.store.save(self.path)


def get_backend(args):
    if args.server:
        from ..store.client import StoreClient
        return StoreClient(args.server)
    return StateBackend(args.state)


def _flush(backend):
    if isinstance(backend, StateBackend):
        backend.flush()


def _fmt_table(rows, headers):
    if not rows:
        widths = [len(h) for h in headers]
    else:
        widths = [max(len(str(r[i])) for r in rows + [headers])
                  for i in range(len(headers))]
    line = "  ".join(h.ljust(w) for h, w in zip(headers, widths))
    out = [line]
    for r in rows:
        out.append("  ".join(str(c).ljust(w) for c, w in zip(r, widths)))
    return "\n".join(out)


# -- job commands -------------------------------------------------------------

def job_run(backend, args):
    spec = JobSpec(
        queue=args.queue,
        min_available=args.min_available,
        tasks=[TaskSpec(name="default", replicas=args.replicas,
                        template={"resources": {"cpu": args.cpu,
                                                "memory": args.mem}})])
    if args.filename:
        with open(args.filename) as f:
            data = yaml.safe_load(f)
        job = from_dict(Job, data)
        if args.name:
            job.meta.name = args.name
    else:
        job = Job(meta=ObjectMeta(name=args.name, namespace=args.namespace),
                  spec=spec)
    backend.create("Job", job)
    _flush(backend)
    print(f"job.batch.volcano.sh/{job.meta.name} created")


def job_list(backend, args):
    rows = []
    for j in backend.list("Job", namespace=args.namespace or None):
        st = j.status
        rows.append((j.meta.name, st.phase, st.pending, st.running,
                     st.succeeded, st.failed, j.spec.queue))
    print(_fmt_table(rows, ["NAME", "PHASE", "PENDING", "RUNNING",
                            "SUCCEEDED", "FAILED", "QUEUE"]))


def job_view(backend, args):
    j = backend.get("Job", args.namespace, args.name)
    if j is None:
        print(f"job {args.name} not found", file=sys.stderr)
        sys.exit(1)
    print(yaml.safe_dump(to_dict(j), sort_keys=False))


def job_delete(backend, args):
    backend.delete("Job", args.namespace, args.name)
    _flush(backend)
    print(f"job.batch.volcano.sh/{args.name} deleted")


def _job_command(backend, args, action: str, verb: str):
    cmd = Command(meta=ObjectMeta(name=f"{verb}-{args.name}",
                                  namespace=args.namespace),
                  action=action, target_kind="Job", target_name=args.name)
    backend.create("Command", cmd)
    _flush(backend)
    print(f"command {verb} issued for job {args.name}")


def job_suspend(backend, args):
    _job_command(backend, args, "AbortJob", "suspend")


def job_resume(backend, args):
    _job_command(backend, args, "ResumeJob", "resume")


# -- queue commands -----------------------------------------------------------

def queue_create(backend, args):
    q = Queue(meta=ObjectMeta(name=args.name),
              spec=QueueSpec(weight=args.weight,
                             capability=Resource.from_spec(
                                 dict(kv.split("=") for kv in args.capability))
                             if args.capability else Resource(),
                             reclaimable=not args.no_reclaimable,
                             parent=args.parent or ""))
    backend.create("Queue", q)
    _flush(backend)
    print(f"queue.scheduling.volcano.sh/{args.name} created")


def queue_list(backend, args):
    rows = []
    for q in backend.list("Queue"):
        rows.append((q.meta.name, q.spec.weight, q.status.state,
                     q.status.inqueue, q.status.running, q.spec.parent or "-"))
    print(_fmt_table(rows, ["NAME", "WEIGHT", "STATE", "INQUEUE", "RUNNING",
                            "PARENT"]))


def queue_get(backend, args):
    q = backend.get("Queue", "default", args.name)
    if q is None:
        print(f"queue {args.name} not found", file=sys.stderr)
        sys.exit(1)
    print(yaml.safe_dump(to_dict(q), sort_keys=False))


def queue_delete(backend, args):
    backend.delete("Queue", "default", args.name)
    _flush(backend)
    print(f"queue.scheduling.volcano.sh/{args.name} deleted")


def queue_operate(backend, args):
    action = "CloseQueue" if args.close else "OpenQueue"
    cmd = Command(meta=ObjectMeta(name=f"queue-op-{args.name}"),
                  action=action, target_kind="Queue", target_name=args.name)
    backend.create("Command", cmd)
    _flush(backend)
    print(f"command {action} issued for queue {args.name}")


# -- pod / jobflow ------------------------------------------------------------

def pod_list(backend, args):
    rows = []
    for p in backend.list("Pod", namespace=args.namespace or None):
        rows.append((p.meta.name, p.phase, p.node_name or "-",
                     p.podgroup_name or "-"))
    print(_fmt_table(rows, ["NAME", "PHASE", "NODE", "PODGROUP"]))


def podgroup_list(backend, args):
    """reference pkg/cli/podgroup list-podgroup.go"""
    rows = []
    for g in backend.list("PodGroup", namespace=args.namespace or None):
        rows.append((g.meta.namespace, g.meta.name, g.spec.queue,
                     g.spec.min_member, g.status.phase))
    print(_fmt_table(rows, ["NAMESPACE", "NAME", "QUEUE", "MINMEMBER",
                            "PHASE"]))


def jobflow_list(backend, args):
    rows = []
    for f in backend.list("JobFlow"):
        rows.append((f.meta.name, f.status.get("state", "-"),
                     len(f.flows)))
    print(_fmt_table(rows, ["NAME", "STATE", "STEPS"]))


def jobtemplate_list(backend, args):
    rows = []
    for t in backend.list("JobTemplate"):
        rows.append((t.meta.name, len(t.spec.tasks),
                     t.spec.total_replicas))
    print(_fmt_table(rows, ["NAME", "TASKS", "REPLICAS"]))


def hypernode_list(backend, args):
    rows = []
    for h in backend.list("HyperNode"):
        members = sum(len(m.selector.exact_match) for m in h.members)
        rows.append((h.meta.name, h.tier, members))
    print(_fmt_table(rows, ["NAME", "TIER", "MEMBERS"]))


def node_list(backend, args):
    rows = []
    for n in backend.list("Node"):
        rows.append((n.meta.name,
                     f"{n.allocatable.milli_cpu / 1000:g}",
                     f"{n.allocatable.memory / (1024 ** 3):g}Gi",
                     "Ready" if n.ready else "NotReady",
                     "unschedulable" if n.unschedulable else "-"))
    print(_fmt_table(rows, ["NAME", "CPU", "MEMORY", "STATUS", "TAINT"]))


def build_parser() -> argparse.ArgumentParser:
    ap = argparse.ArgumentParser(prog="vcctl",
                                 description="volcano_amd CLI")
    ap.add_argument("--server", default=None,
                    help="apiserver URL (e.g. http://127.0.0.1:8343)")
    ap.add_argument("--state", default="/tmp/volcano-amd-state.json",
                    help="state file when no --server")
    sub = ap.add_subparsers(dest="group", required=True)

    job = sub.add_parser("job").add_subparsers(dest="cmd", required=True)
    run = job.add_parser("run")
    run.add_argument("--name", "-N", required=True)
    run.add_argument("--namespace", "-n", default="default")
    run.add_argument("--replicas", "-r", type=int, default=1)
    run.add_argument("--min-available", "-m", type=int, default=None)
    run.add_argument("--queue", "-q", default="default")
    run.add_argument("--cpu", default="1")
    run.add_argument("--mem", default="1Gi")
    run.add_argument("--filename", "-f", default=None)
    run.set_defaults(fn=job_run)
    for verb, fn in [("list", job_list)]:
        p = job.add_parser(verb)
        p.add_argument("--namespace", "-n", default=None)
        p.set_defaults(fn=fn)
    for verb, fn in [("view", job_view), ("delete", job_delete),
                     ("suspend", job_suspend), ("resume", job_resume)]:
        p = job.add_parser(verb)
        p.add_argument("--name", "-N", required=True)
        p.add_argument("--namespace", "-n", default="default")
        p.set_defaults(fn=fn)

    queue = sub.add_parser("queue").add_subparsers(dest="cmd", required=True)
    qc = queue.add_parser("create")
    qc.add_argument("--name", "-N", required=True)
    qc.add_argument("--weight", "-w", type=int, default=1)
    qc.add_argument("--capability", "-c", nargs="*", default=None,
                    help="e.g. cpu=64 memory=128Gi")
    qc.add_argument("--no-reclaimable", action="store_true")
    qc.add_argument("--parent", default=None)
    qc.set_defaults(fn=queue_create)
    queue.add_parser("list").set_defaults(fn=queue_list)
    for verb, fn in [("get", queue_get), ("delete", queue_delete)]:
        p = queue.add_parser(verb)
        p.add_argument("--name", "-N", required=True)
        p.set_defaults(fn=fn)
    op = queue.add_parser("operate")
    op.add_argument("--name", "-N", required=True)
    op.add_argument("--close", action="store_true")
    op.set_defaults(fn=queue_operate)

    pod = sub.add_parser("pod").add_subparsers(dest="cmd", required=True)
    pl = pod.add_parser("list")
    pl.add_argument("--namespace", "-n", default=None)
    pl.set_defaults(fn=pod_list)

    pgp = sub.add_parser("podgroup").add_subparsers(dest="cmd",
                                                    required=True)
    pgl = pgp.add_parser("list")
    pgl.add_argument("--namespace", "-n", default=None)
    pgl.set_defaults(fn=podgroup_list)

    jf = sub.add_parser("jobflow").add_subparsers(dest="cmd", required=True)
    jf.add_parser("list").set_defaults(fn=jobflow_list)

    jt = sub.add_parser("jobtemplate").add_subparsers(dest="cmd",
                                                      required=True)
    jt.add_parser("list").set_defaults(fn=jobtemplate_list)

    hn = sub.add_parser("hypernode").add_subparsers(dest="cmd",
                                                    required=True)
    hn.add_parser("list").set_defaults(fn=hypernode_list)

    nd = sub.add_parser("node").add_subparsers(dest="cmd", required=True)
    nd.add_parser("list").set_defaults(fn=node_list)

    return ap


def main(argv=None) -> int:
    args = build_parser().parse_args(argv)
    backend = get_backend(args)
    try:
        args.fn(backend, args)
    except Exception as e:
        print(f"error: {e}", file=sys.stderr)
        return 1
    return 0


if __name__ == "__main__":
    sys.exit(main())
