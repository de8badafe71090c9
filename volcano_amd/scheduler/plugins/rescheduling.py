"""Rescheduling plugin (reference ``plugins/rescheduling``): periodic
re-balancing — strategies nominate running tasks to move; the shuffle
action evicts them.  Strategy implemented: ``lowNodeUtilization``
(evacuate nodes whose requested utilization sits below thresholds so
they can be drained/binpacked)."""

from __future__ import annotations

import time

from ...api.resource import CPU, MEMORY
from ...api.types import TaskStatus
from .base import Plugin, register


@register("rescheduling")
class ReschedulingPlugin(Plugin):
    _last_run = 0.0

    def on_session_open(self, ssn) -> None:
        interval = float(self.args.get("interval", 300.0))
        strategy = self.args.get("strategies", "lowNodeUtilization")
        thresholds = self.args.get("lowNodeUtilization", {}) or {}
        cpu_t = float(thresholds.get("cpu", 20.0))
        mem_t = float(thresholds.get("memory", 20.0))

        now = time.time()
        if now - ReschedulingPlugin._last_run < interval:
            return
        if "lowNodeUtilization" not in str(strategy):
            return

        def victim_tasks(tasks):
            ReschedulingPlugin._last_run = time.time()
            victims = []
            for ni in ssn.nodes.values():
                alloc = ni.allocatable
                if alloc.get(CPU) <= 0:
                    continue
                cpu_pct = 100.0 * ni.used.get(CPU) / max(alloc.get(CPU), 1.0)
                mem_pct = 100.0 * ni.used.get(MEMORY) / max(alloc.get(MEMORY), 1.0)
                if 0 < cpu_pct < cpu_t and mem_pct < mem_t:
                    victims.extend(
                        t for t in ni.tasks.values()
                        if t.status in (TaskStatus.RUNNING, TaskStatus.BOUND)
                        and t.preemptable)
            return victims

        ssn.victim_tasks_fns.append(victim_tasks)
