"""Priority plugin (reference ``plugins/priority``): order jobs/tasks by
priority; victims must have lower priority than the preemptor."""

from __future__ import annotations

from .base import Plugin, register


@register("priority")
class PriorityPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        from ...utils.features import enabled
        if not enabled("PriorityClass"):
            return
        def job_order(a, b) -> int:
            if a.priority != b.priority:
                return -1 if a.priority > b.priority else 1
            return 0

        def task_order(a, b) -> int:
            if a.priority != b.priority:
                return -1 if a.priority > b.priority else 1
            return 0

        def preemptable(preemptor, candidates):
            return [v for v in candidates if v.priority < preemptor.priority]

        def job_starving(job) -> bool:
            return job.is_starving()

        ssn.add_job_order_fn(job_order, key=lambda j: -j.priority,
                             col=lambda jt, rows: -jt.prio[rows])
        ssn.add_task_order_fn(task_order)
        ssn.preemptable_fns.append(preemptable)
        ssn.job_starving_fns.append(job_starving)
