"""Garbage collector (reference ``pkg/controllers/garbagecollector/``):
deletes finished Jobs after ttlSecondsAfterFinished."""

from __future__ import annotations

import time

from ..api.types import JobPhase
from .framework import Controller, register

FINISHED = (JobPhase.COMPLETED.value, JobPhase.FAILED.value,
            JobPhase.TERMINATED.value, JobPhase.ABORTED.value)


@register("garbagecollector")
class GarbageCollector(Controller):
    watch_kinds = ()

    def resync(self) -> None:
        now = time.time()
        for job in self.store.list("Job"):
            ttl = job.spec.ttl_seconds_after_finished
            if ttl is None or job.status.phase not in FINISHED:
                continue
            finished_at = job.meta.annotations.get("volcano.sh/finished-at")
            if finished_at is None:
                job.meta.annotations["volcano.sh/finished-at"] = str(now)
                self.store.update("Job", job)
                continue
            if now - float(finished_at) >= ttl:
                self.store.delete("Job", job.meta.namespace, job.meta.name)
