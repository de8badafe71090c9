"""JobTemplate controller (reference ``pkg/controllers/jobtemplate``):
standalone bookkeeping — tracks which Jobs were created from each
template (status.jobDependsOnList) so flows and users can see template
fan-out; templates themselves create nothing."""

from __future__ import annotations

from ..store import EventType
from .framework import Controller, register

LBL_TEMPLATE = "flow.volcano.sh/jobtemplate"


@register("jobtemplate")
class JobTemplateController(Controller):
    watch_kinds = ("JobTemplate", "Job")

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty: set = set()

    def handle(self, ev) -> None:
        if ev.kind == "JobTemplate":
            if ev.type != EventType.DELETED:
                self._dirty.add(ev.obj.meta.key)
        elif ev.kind == "Job":
            tmpl = ev.obj.meta.labels.get(LBL_TEMPLATE)
            if tmpl:
                self._dirty.add(f"{ev.obj.meta.namespace}/{tmpl}")

    def resync(self) -> None:
        dirty, self._dirty = self._dirty, set()
        for key in dirty:
            ns, name = key.split("/", 1)
            tmpl = self.store.get("JobTemplate", ns, name)
            if tmpl is None:
                continue
            jobs = sorted(
                j.meta.name for j in self.store.list("Job", namespace=ns)
                if j.meta.labels.get(LBL_TEMPLATE) == name)
            if tmpl.status.get("jobDependsOnList") != jobs:
                tmpl.status["jobDependsOnList"] = jobs
                self.store.update("JobTemplate", tmpl)
