"""Rescheduling plugin (reference ``plugins/rescheduling``): periodic
re-balancing — strategies nominate running tasks to move; the shuffle
action evicts them.  Strategies (the reference registry
rescheduling.go:45-61 is extensible, with lowNodeUtilization in-tree):

* ``lowNodeUtilization`` — evacuate nodes whose requested utilization
  sits below thresholds so they can be drained/binpacked;
* ``highNodeUtilization`` — relieve overloaded nodes by moving their
  offline (preemptable) pods elsewhere.
"""

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

        high = self.args.get("highNodeUtilization", {}) or {}
        cpu_hi = float(high.get("cpu", 85.0))
        mem_hi = float(high.get("memory", 85.0))

        now = time.time()
        if now - ReschedulingPlugin._last_run < interval:
            return
        strategies = [x.strip() for x in str(strategy).split(",")]
        if not any(st in ("lowNodeUtilization", "highNodeUtilization")
                   for st in strategies):
            return

        def victim_tasks(tasks):
            ReschedulingPlugin._last_run = time.time()
            victims = []
            for ni in ssn.nodes.values():
                alloc = ni.allocatable
                if alloc.get(CPU) <= 0:
                    continue
                cpu_pct = 100.0 * ni.used.get(CPU) / max(alloc.get(CPU), 1.0)
                mem_pct = 100.0 * ni.used.get(MEMORY) / \
                    max(alloc.get(MEMORY), 1.0)
                take = False
                if "lowNodeUtilization" in strategies and \
                        0 < cpu_pct < cpu_t and mem_pct < mem_t:
                    take = True
                if "highNodeUtilization" in strategies and \
                        (cpu_pct > cpu_hi or mem_pct > mem_hi):
                    take = True
                if take:
                    victims.extend(
                        t for t in ni.tasks.values()
                        if t.status in (TaskStatus.RUNNING, TaskStatus.BOUND)
                        and t.preemptable)
            return victims

        ssn.victim_tasks_fns.append(victim_tasks)
