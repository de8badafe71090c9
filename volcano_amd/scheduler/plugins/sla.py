"""SLA plugin (reference ``plugins/sla/sla.go:134-153``): job waiting-time
guarantee — a job past its ``sla-waiting-time`` jumps the order and must
be admitted."""

from __future__ import annotations

import time

from ..session import ABSTAIN, PERMIT
from .base import Plugin, register


def _parse_duration(s) -> float:
    if isinstance(s, (int, float)):
        return float(s)
    s = str(s).strip()
    units = {"s": 1.0, "m": 60.0, "h": 3600.0}
    if s and s[-1] in units:
        return float(s[:-1]) * units[s[-1]]
    return float(s)


@register("sla")
class SlaPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        default_wait = self.args.get("sla-waiting-time")
        if default_wait is None:
            return
        wait = _parse_duration(default_wait)
        now = time.time()

        def overdue(job) -> bool:
            ann = (job.podgroup.meta.annotations.get("sla-waiting-time")
                   if job.podgroup else None)
            w = _parse_duration(ann) if ann else wait
            return (now - job.creation_timestamp) > w

        def job_order(a, b) -> int:
            oa, ob = overdue(a), overdue(b)
            if oa != ob:
                return -1 if oa else 1
            return 0

        def job_enqueueable(job) -> int:
            return PERMIT if overdue(job) else ABSTAIN

        def job_pipelined(job) -> int:
            return PERMIT if overdue(job) else ABSTAIN

        ssn.add_job_order_fn(job_order, key=lambda j: not overdue(j))
        ssn.job_enqueueable_fns.append(job_enqueueable)
        ssn.job_pipelined_fns.append(job_pipelined)
