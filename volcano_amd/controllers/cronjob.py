"""CronJob controller (reference ``pkg/controllers/cronjob/``): creates
vcjobs on a cron schedule, honoring concurrencyPolicy
(Allow/Forbid/Replace), startingDeadlineSeconds and history limits."""

from __future__ import annotations

import time
from typing import List

from ..api.objects import Job, ObjectMeta
from ..api.types import JobPhase
from ..utils.cron import CronSchedule
from .framework import Controller, register
from .garbagecollector import FINISHED


@register("cronjob")
class CronJobController(Controller):
    watch_kinds = ()

    def initialize(self, store) -> None:
        super().initialize(store)
        self.now = None     # test hook: override the clock

    def _time(self) -> float:
        return self.now if self.now is not None else time.time()

    def _children(self, cj) -> List[Job]:
        owner = f"CronJob/{cj.meta.key}"
        return self.store.list("Job", namespace=cj.meta.namespace,
                               selector=lambda j: j.meta.owner == owner)

    def resync(self) -> None:
        from ..utils.features import enabled
        if not enabled("CronVolcanoJobSupport"):
            return

        for cj in self.store.list("CronJob"):
            self._sync(cj)

    def _sync(self, cj) -> None:
        """reference cronjob_controller.go:172-219 syncCronJob."""
        if cj.suspend:
            return
        now = self._time()
        last = cj.status.get("last_schedule_time", cj.meta.creation_timestamp)
        sched = CronSchedule(cj.schedule)
        nxt = sched.next_after(last)
        if nxt is None or nxt > now:
            return
        if cj.starting_deadline_seconds is not None and \
                now - nxt > cj.starting_deadline_seconds:
            cj.status["last_schedule_time"] = now    # missed the window
            self.store.update("CronJob", cj)
            return

        children = self._children(cj)
        active = [j for j in children if j.status.phase not in FINISHED]
        if active:
            if cj.concurrency_policy == "Forbid":
                cj.status["last_schedule_time"] = nxt
                self.store.update("CronJob", cj)
                return
            if cj.concurrency_policy == "Replace":
                for j in active:
                    self.store.delete("Job", j.meta.namespace, j.meta.name)

        name = f"{cj.meta.name}-{int(nxt) // 60}"
        if self.store.get("Job", cj.meta.namespace, name) is None:
            self.store.create("Job", Job(
                meta=ObjectMeta(name=name, namespace=cj.meta.namespace,
                                owner=f"CronJob/{cj.meta.key}"),
                spec=cj.job_template))
        cj.status["last_schedule_time"] = nxt
        self.store.update("CronJob", cj)

        # history limits
        done = sorted((j for j in self._children(cj)
                       if j.status.phase in FINISHED),
                      key=lambda j: j.meta.creation_timestamp)
        ok = [j for j in done if j.status.phase == JobPhase.COMPLETED.value]
        bad = [j for j in done if j.status.phase != JobPhase.COMPLETED.value]
        for j in ok[:-cj.successful_jobs_history_limit or None]:
            self.store.delete("Job", j.meta.namespace, j.meta.name)
        for j in bad[:-cj.failed_jobs_history_limit or None]:
            self.store.delete("Job", j.meta.namespace, j.meta.name)
