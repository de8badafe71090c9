"""HyperJob controller (reference ``apis/training/v1alpha1`` HyperJob —
multi-cluster job splitting, incubating): expands a HyperJob into
``replicas`` member vcjobs and aggregates their phases."""

from __future__ import annotations

from ..api.objects import HyperJob, Job, ObjectMeta
from ..api.types import JobPhase
from .framework import Controller, register


@register("hyperjob")
class HyperJobController(Controller):
    watch_kinds = ("HyperJob", "Job")

    def handle(self, ev) -> None:
        pass

    def resync(self) -> None:
        for hj in self.store.list("HyperJob"):
            self._sync(hj)

    def _member_name(self, hj: HyperJob, i: int) -> str:
        return f"{hj.meta.name}-{i}"

    def _sync(self, hj: HyperJob) -> None:
        import copy
        phases = []
        for i in range(hj.replicas):
            name = self._member_name(hj, i)
            job = self.store.get("Job", hj.meta.namespace, name)
            if job is None:
                job = Job(meta=ObjectMeta(name=name,
                                          namespace=hj.meta.namespace,
                                          owner=f"HyperJob/{hj.meta.key}"),
                          spec=copy.deepcopy(hj.job_template))
                self.store.create("Job", job)
            phases.append(job.status.phase)

        if all(p == JobPhase.COMPLETED.value for p in phases):
            state = "Completed"
        elif any(p in (JobPhase.FAILED.value, JobPhase.ABORTED.value)
                 for p in phases):
            state = "Failed"
        elif any(p == JobPhase.RUNNING.value for p in phases):
            state = "Running"
        else:
            state = "Pending"
        if hj.status.get("state") != state:
            hj.status["state"] = state
            hj.status["members"] = {self._member_name(hj, i): p
                                    for i, p in enumerate(phases)}
            self.store.update("HyperJob", hj)
