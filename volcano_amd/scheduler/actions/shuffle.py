"""Shuffle action (reference ``actions/shuffle/shuffle.go:48-71``).

Rescheduling: collect all running tasks, ask the VictimTasks plugins
(rescheduling/tdm) which should move, evict them — the job controller /
next cycles re-place them.
"""

from __future__ import annotations

from ...api.types import TaskStatus
from ..statement import Statement


class ShuffleAction:
    name = "shuffle"

    def execute(self, ssn) -> None:
        if not ssn.victim_tasks_fns:
            return
        running = []
        for job in ssn.jobs.values():
            running.extend(job.tasks_with_status(TaskStatus.RUNNING,
                                                 TaskStatus.BOUND))
        victims = ssn.victim_tasks(running)
        if not victims:
            return
        stmt = Statement(ssn)
        for v in victims:
            stmt.evict(v, reason="shuffle")
        stmt.commit()
