"""Admission handlers (reference ``pkg/webhooks/admission/``: jobs
mutate/validate, queues mutate/validate, podgroups, pods, jobflows,
cronjobs — SURVEY §2.5).
"""

from __future__ import annotations

from ..api.objects import DEFAULT_QUEUE
from ..api.types import Action, Event, QueueState
from ..utils.cron import CronSchedule
from .router import AdmissionChain, AdmissionError, AdmissionService

VALID_EVENTS = {e.value for e in Event}
VALID_ACTIONS = {a.value for a in Action}
VALID_POLICY_EVENTS = VALID_EVENTS | {"*"}


# -- jobs ---------------------------------------------------------------------

def mutate_job(store, job, op) -> None:
    """Defaulting (reference admission/jobs/mutate): queue, task names,
    minAvailable."""
    if not job.spec.queue:
        job.spec.queue = DEFAULT_QUEUE
    for i, ts in enumerate(job.spec.tasks):
        if not ts.name:
            ts.name = f"task-{i}"
        if ts.replicas < 0:
            ts.replicas = 0
    if job.spec.min_available is None:
        job.spec.min_available = sum(t.min_needed for t in job.spec.tasks)


_DNS1123 = None


def _dns1123(name: str) -> bool:
    global _DNS1123
    if _DNS1123 is None:
        import re
        _DNS1123 = re.compile(r"^[a-z0-9]([-a-z0-9]*[a-z0-9])?$")
    return bool(_DNS1123.fullmatch(name)) and len(name) <= 63


def validate_job(store, job, op) -> None:
    """reference admission/jobs/validate (admit_job.go:120-230): spec
    legality incl. plugin names, MPI master presence, policy sets,
    networkTopology, task DAG, pod-name length and leaf-queue rules."""
    if not job.meta.name:
        raise AdmissionError("job name required")
    if not _dns1123(job.meta.name):
        raise AdmissionError(
            f"job name {job.meta.name!r} must be a DNS-1123 label")
    if not job.spec.tasks:
        raise AdmissionError("No task specified in job spec")
    names = [t.name for t in job.spec.tasks]
    if len(names) != len(set(names)):
        raise AdmissionError(f"duplicated task names: {names}")
    total = job.spec.total_replicas
    if job.spec.min_available is not None:
        if job.spec.min_available < 0:
            raise AdmissionError("minAvailable must be >= 0")
        if job.spec.min_available > total:
            raise AdmissionError(
                f"minAvailable {job.spec.min_available} > "
                f"total replicas {total}")
    if job.spec.min_success is not None and job.spec.min_success < 0:
        raise AdmissionError("minSuccess must be >= 0")
    if job.spec.ttl_seconds_after_finished is not None \
            and job.spec.ttl_seconds_after_finished < 0:
        raise AdmissionError("ttlSecondsAfterFinished must be >= 0")
    for i, ts in enumerate(job.spec.tasks):
        if ts.name and not _dns1123(ts.name):
            raise AdmissionError(
                f"task name {ts.name!r} must be a DNS-1123 label")
        # generated pod names must stay within the k8s 253-char bound
        if len(f"{job.meta.name}-{ts.name}-{ts.replicas}") > 253:
            raise AdmissionError(f"task {ts.name}: pod name too long")
        if ts.replicas < 0:
            raise AdmissionError(f"task {ts.name}: replicas must be >= 0")
        if ts.min_available is not None and ts.min_available > ts.replicas:
            raise AdmissionError(
                f"task {ts.name}: minAvailable > replicas")
        _validate_policy_set(ts.policies, f"task {ts.name}")
    _validate_policy_set(job.spec.policies, "job")
    if job.spec.max_retry < 0:
        raise AdmissionError("maxRetry must be >= 0")
    _validate_network_topology(job.spec.network_topology)
    # job plugins must be registered (admit_job.go "unable to find job
    # plugin"); the MPI plugin additionally needs its master task present
    if job.spec.plugins:
        from ..controllers.jobplugins import JOB_PLUGINS
        for pname in job.spec.plugins:
            if pname not in JOB_PLUGINS:
                raise AdmissionError(f"unable to find job plugin: {pname}")
        if "mpi" in job.spec.plugins:
            # reference: the configured mpi master task must exist
            # (admit_job.go:134-141); enforced when explicitly configured
            args = job.spec.plugins.get("mpi") or []
            master = next((a.split("=", 1)[1] for a in args
                           if a.startswith("master=")), None)
            if master is not None and \
                    all(t.name != master for t in job.spec.tasks):
                raise AdmissionError(
                    "the specified mpi master task was not found")
    # task dependsOn must form a DAG (admit_job.go topoSort)
    if any(ts.depends_on for ts in job.spec.tasks):
        _validate_task_dag(job.spec.tasks)
    if store is not None and job.spec.queue:
        q = store.get("Queue", "default", job.spec.queue)
        if q is None:
            raise AdmissionError(f"queue {job.spec.queue!r} does not exist")
        if q.status.state != QueueState.OPEN.value:
            raise AdmissionError(f"queue {job.spec.queue!r} is not open")
        if q.meta.name == "root":
            raise AdmissionError("can not submit job to root queue")
        children = [c for c in store.list("Queue")
                    if c.spec.parent == q.meta.name]
        if children:
            raise AdmissionError(
                f"can only submit job to leaf queue; queue "
                f"{q.meta.name!r} has {len(children)} child queues")


def _validate_network_topology(spec) -> None:
    if not spec:
        return
    if spec.get("mode", "hard") not in ("hard", "soft"):
        raise AdmissionError("networkTopology.mode must be hard|soft")
    t = spec.get("highestTierAllowed")
    if t is not None and int(t) < 1:
        raise AdmissionError("networkTopology.highestTierAllowed must be >= 1")


def _validate_task_dag(tasks) -> None:
    names = {t.name for t in tasks}
    state = {}

    def visit(n):
        if state.get(n) == 1:
            raise AdmissionError(
                "job task dependencies do not form a DAG")
        if state.get(n) == 2:
            return
        state[n] = 1
        ts = next(t for t in tasks if t.name == n)
        for d in ts.depends_on:
            if d not in names:
                raise AdmissionError(
                    f"task {n}: unknown dependency {d!r}")
            visit(d)
        state[n] = 2

    for t in tasks:
        visit(t.name)


def _validate_policy_set(policies, where: str) -> None:
    """validatePolicies (admit_job.go): per-policy legality + duplicate
    events + '*' exclusivity."""
    seen = set()
    for pol in policies:
        _validate_policy(pol, where)
        evs = pol.events or ([pol.event] if pol.event else [])
        for e in evs:
            if e in seen:
                raise AdmissionError(f"{where}: duplicated policy event {e}")
            seen.add(e)
    if "*" in seen and len(seen) > 1:
        raise AdmissionError(
            f"{where}: '*' policy cannot be combined with other events")


def _validate_policy(pol, where: str) -> None:
    evs = pol.events or ([pol.event] if pol.event else [])
    for e in evs:
        if e not in VALID_POLICY_EVENTS:
            raise AdmissionError(f"{where}: invalid policy event {e!r}")
    if pol.action and pol.action not in VALID_ACTIONS:
        raise AdmissionError(f"{where}: invalid policy action {pol.action!r}")


# -- queues -------------------------------------------------------------------

def mutate_queue(store, queue, op) -> None:
    if queue.spec.weight <= 0:
        queue.spec.weight = 1


def validate_queue(store, queue, op) -> None:
    if not queue.meta.name:
        raise AdmissionError("queue name required")
    if op == "DELETE":
        # validate_queue.go validateQueueDeleting: default/root are
        # permanent; parents with children must be drained bottom-up
        if queue.meta.name in ("default", "root"):
            raise AdmissionError(
                f"`{queue.meta.name}` queue can not be deleted")
        if store is not None:
            kids = [q.meta.name for q in store.list("Queue")
                    if q.spec.parent == queue.meta.name]
            if kids:
                raise AdmissionError(
                    f"queue {queue.meta.name} can not be deleted because "
                    f"it has {len(kids)} child queues: {', '.join(kids)}")
        return
    # state legality (validateStateOfQueue)
    if queue.status.state not in ("", "Open", "Closed"):
        raise AdmissionError(
            "queue state must be in [Open Closed]")
    # guarantee ≤ deserved ≤ capability, dimension-wise where set
    # (validateResourceQuantityOfQueue)
    cap = queue.spec.capability.q
    des = queue.spec.deserved.q
    gua = queue.spec.guarantee.q
    for r, g in gua.items():
        d = des.get(r)
        if d is None or d < g:
            raise AdmissionError(
                f"deserved[{r}] must be >= guarantee[{r}]={g:g}")
    for r, d in des.items():
        c = cap.get(r)
        if c is not None and c < d:
            raise AdmissionError(
                f"capability[{r}]={c:g} must be >= deserved[{r}]={d:g}")
    # hierarchy annotations: path and weights must align, weights > 0
    # (validate_queue.go:150-176)
    h = queue.meta.annotations.get("volcano.sh/hierarchy", "")
    hw = queue.meta.annotations.get("volcano.sh/hierarchy-weights", "")
    if h or hw:
        paths = h.split("/")
        weights = hw.split("/")
        if len(paths) != len(weights):
            raise AdmissionError(
                f"hierarchy {h!r} and weights {hw!r} must have the "
                "same depth")
        for w in weights:
            try:
                ok = float(w) > 0
            except ValueError:
                ok = False
            if not ok:
                raise AdmissionError(
                    f"hierarchy weight {w!r} must be a positive number")
    if queue.spec.parent and store is not None:
        parent = store.get("Queue", "default", queue.spec.parent)
        if parent is None:
            raise AdmissionError(
                f"parent queue {queue.spec.parent!r} does not exist")
        # hierarchy legality: no cycles up the chain
        seen = {queue.meta.name}
        p = parent
        while p is not None:
            if p.meta.name in seen:
                raise AdmissionError("queue hierarchy cycle")
            seen.add(p.meta.name)
            p = store.get("Queue", "default", p.spec.parent) \
                if p.spec.parent else None


# -- podgroups ----------------------------------------------------------------

def mutate_podgroup(store, pg, op) -> None:
    """reference mutate_podgroup.go:104-125: a podgroup still on the
    default queue inherits its namespace's queue-name annotation."""
    if pg.spec.queue != DEFAULT_QUEUE or store is None:
        return
    ns = store.get("Namespace", "", pg.meta.namespace) or         store.get("Namespace", "default", pg.meta.namespace)
    if ns is None:
        return
    q = ns.meta.annotations.get("scheduling.volcano.sh/queue-name")
    if q:
        pg.spec.queue = q


def validate_podgroup(store, pg, op) -> None:
    if pg.spec.min_member < 0:
        raise AdmissionError("minMember must be >= 0")
    for role, n in pg.spec.min_task_member.items():
        if n < 0:
            raise AdmissionError(f"minTaskMember[{role}] must be >= 0")
    for i, pol in enumerate(pg.spec.sub_group_policy or []):
        if int(pol.get("subGroupSize", 1)) < 1:
            raise AdmissionError(
                f"subGroupPolicy[{i}].subGroupSize must be >= 1")
        if int(pol.get("minSubGroups", 0)) < 0:
            raise AdmissionError(
                f"subGroupPolicy[{i}].minSubGroups must be >= 0")
        nt = pol.get("networkTopology")
        if nt is not None and nt.get("mode", "hard") not in ("hard", "soft"):
            raise AdmissionError(
                f"subGroupPolicy[{i}].networkTopology.mode must be "
                "hard|soft")


# -- pods ---------------------------------------------------------------------

JDB_MIN_AVAILABLE = "scheduling.volcano.sh/jdb-min-available"
JDB_MAX_UNAVAILABLE = "scheduling.volcano.sh/jdb-max-unavailable"


def _int_or_percentage(key: str, value: str) -> None:
    """validateIntPercentageStr (admit_pod.go): "3" or "25%", > 0."""
    v = value.strip()
    if v.endswith("%"):
        v = v[:-1]
    try:
        ok = int(v) > 0
    except ValueError:
        ok = False
    if not ok:
        raise AdmissionError(
            f"invalid value {value!r} for annotation {key}: must be a "
            "positive integer or percentage")


def validate_pod(store, pod, op) -> None:
    """reference admission/pods/validate/admit_pod.go:99-135 — the job
    disruption budget annotations must parse and are mutually
    exclusive."""
    anns = pod.meta.annotations
    present = [k for k in (JDB_MIN_AVAILABLE, JDB_MAX_UNAVAILABLE)
               if k in anns]
    for k in present:
        _int_or_percentage(k, anns[k])
    if len(present) > 1:
        raise AdmissionError(
            f"not allow configure multiple annotations {present} at "
            "same time")


def mutate_pod(store, pod, op) -> None:
    """reference admission/pods/mutate: annotate the scheduler."""
    if not pod.scheduler_name:
        pod.scheduler_name = "volcano"


# -- jobflows -----------------------------------------------------------------

def validate_jobflow(store, flow, op) -> None:
    names = {s.name for s in flow.flows}
    if len(names) != len(flow.flows):
        raise AdmissionError("duplicated flow step names")
    for s in flow.flows:
        for d in s.depends_on:
            if d not in names:
                raise AdmissionError(
                    f"step {s.name}: unknown dependency {d!r}")
    # cycle check
    state = {}

    def visit(n):
        if state.get(n) == 1:
            raise AdmissionError("cycle in dependsOn graph")
        if state.get(n) == 2:
            return
        state[n] = 1
        step = next(s for s in flow.flows if s.name == n)
        for d in step.depends_on:
            visit(d)
        state[n] = 2

    for s in flow.flows:
        visit(s.name)


# -- cronjobs -----------------------------------------------------------------

def validate_hypernode(store, hn, op) -> None:
    """reference admission/hypernodes/validate: regex selectors must
    compile; tier >= 1; every member needs a selector."""
    import re
    if hn.tier < 1:
        raise AdmissionError("hypernode tier must be >= 1")
    for i, m in enumerate(hn.members or []):
        sel = m.selector
        if not (sel.exact_match or sel.regex_match or sel.label_match):
            raise AdmissionError(f"members[{i}]: empty selector")
        if sel.regex_match:
            try:
                re.compile(sel.regex_match)
            except re.error as e:
                raise AdmissionError(
                    f"members[{i}]: bad regexMatch pattern: {e}")


def validate_cronjob(store, cj, op) -> None:
    try:
        CronSchedule(cj.schedule)
    except Exception as e:
        raise AdmissionError(f"invalid cron schedule {cj.schedule!r}: {e}")
    if cj.concurrency_policy not in ("Allow", "Forbid", "Replace"):
        raise AdmissionError(
            f"invalid concurrencyPolicy {cj.concurrency_policy!r}")


def register_all(chain: AdmissionChain) -> None:
    """reference webhooks/router/admission.go:32-53 RegisterAdmission."""
    for svc in [
        AdmissionService("/jobs/mutate", "Job", mutate_job),
        AdmissionService("/jobs/validate", "Job", validate_job),
        AdmissionService("/queues/mutate", "Queue", mutate_queue),
        AdmissionService("/queues/validate", "Queue", validate_queue,
                         operations=("CREATE", "UPDATE", "DELETE")),
        AdmissionService("/podgroups/mutate", "PodGroup", mutate_podgroup),
        AdmissionService("/podgroups/validate", "PodGroup", validate_podgroup),
        AdmissionService("/pods/mutate", "Pod", mutate_pod),
        AdmissionService("/pods/validate", "Pod", validate_pod),
        AdmissionService("/jobflows/validate", "JobFlow", validate_jobflow),
        AdmissionService("/cronjobs/validate", "CronJob", validate_cronjob),
        AdmissionService("/hypernodes/validate", "HyperNode",
                         validate_hypernode),
    ]:
        chain.register(svc)
