"""Gang plugin (reference ``plugins/gang/gang.go:57-211``).

JobValid: reject jobs whose valid task count can never reach minAvailable.
JobOrder: not-ready jobs first (ready-last, gang.go:163).  JobReady /
JobPipelined: occupied (+waiting) vs minAvailable.  Preemptable: protect
victims' gangs — a victim is only evictable while its job stays at or
above minAvailable (gang.go:95-133).

The actual gang *commit* (all-or-nothing placement) is on-device: the
select/finalize/revert kernels enforce minAvailable inside the cycle
(ops/csrc/scheduler_kernels.hip); this plugin contributes the host-side
order/valid/victim semantics.
"""

from __future__ import annotations

from ...api.types import TaskStatus
from ..session import PERMIT, REJECT
from .base import Plugin, register


@register("gang")
class GangPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        def job_valid(job) -> bool:
            failed = len(job.task_status_index.get(TaskStatus.FAILED, ()))
            return len(job.tasks) - failed >= job.min_available

        def job_order(a, b) -> int:
            ra, rb = a.is_ready(), b.is_ready()
            if ra != rb:
                return 1 if ra else -1     # not-ready first
            return 0

        def job_ready(job) -> bool:
            return job.is_ready() and job.roles_ready()
        job_ready.is_gang = True      # allocate's ready-flip fast path

        def job_pipelined(job) -> int:
            if job.is_pipelined():
                return PERMIT
            return REJECT

        def job_starving(job) -> bool:
            return job.is_starving()

        def preemptable(preemptor, candidates):
            # count survivors per victim job if all currently-listed victims
            # of that job were evicted one by one; allow only down to min
            out = []
            taken = {}
            for v in candidates:
                job = ssn.jobs.get(v.job_key)
                if job is None:
                    out.append(v)
                    continue
                occ = job.occupied_count - taken.get(v.job_key, 0)
                if occ > job.min_available:
                    out.append(v)
                    taken[v.job_key] = taken.get(v.job_key, 0) + 1
            return out

        # bundle actions (gangpreempt/gangreclaim) evict whole gangs, so
        # the keep-above-min filter does not apply to them — they skip
        # fns carrying this marker (actions/gangpreempt.py _victim_ok)
        preemptable.bundle_exempt = True

        ssn.job_valid_fns.append(job_valid)
        ssn.job_valid_cols.append(
            lambda jt: (jt.ntasks - jt.nfailed) >= jt.minav)
        ssn.add_job_order_fn(job_order, key=lambda j: j.is_ready(),
                             col=lambda jt, rows:
                                 jt.occ[rows] >= jt.minav[rows])
        ssn.job_ready_fns.append(job_ready)
        if self.args.get("enabledJobPipelined", True) is not False:
            ssn.job_pipelined_fns.append(job_pipelined)
        ssn.job_starving_fns.append(job_starving)
        # per-function enable flags (reference conf.PluginOption
        # EnabledPreemptable/EnabledReclaimable — default on)
        if self.args.get("enabledPreemptable", True) is not False:
            ssn.preemptable_fns.append(preemptable)
        if self.args.get("enabledReclaimable", True) is not False:
            ssn.reclaimable_fns.append(preemptable)
