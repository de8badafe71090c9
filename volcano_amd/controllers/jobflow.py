"""JobFlow controller (reference ``pkg/controllers/jobflow/``): a DAG of
JobTemplates — each flow step's vcjob is created once every job it
``dependsOn`` has Completed; flow status aggregates; jobRetainPolicy
delete cleans up after the flow succeeds."""

from __future__ import annotations

import copy

from ..api.objects import Job, JobFlow, ObjectMeta
from ..api.types import JobPhase
from .framework import Controller, register


@register("jobflow")
class JobFlowController(Controller):
    watch_kinds = ("JobFlow", "Job")

    def initialize(self, store) -> None:
        super().initialize(store)

    def handle(self, ev) -> None:
        pass    # resync() walks all flows; events just wake the loop

    def _job_name(self, flow: JobFlow, step: str) -> str:
        return f"{flow.meta.name}-{step}"

    def _validate(self, flow: JobFlow) -> bool:
        """DAG cycle check (reference webhooks jobflows/validate)."""
        names = {s.name for s in flow.flows}
        seen, stack = set(), set()

        def visit(n) -> bool:
            if n in stack:
                return False
            if n in seen:
                return True
            stack.add(n)
            step = next(s for s in flow.flows if s.name == n)
            for d in step.depends_on:
                if d in names and not visit(d):
                    return False
            stack.discard(n)
            seen.add(n)
            return True

        return all(visit(s.name) for s in flow.flows)

    def resync(self) -> None:
        for flow in self.store.list("JobFlow"):
            self._sync(flow)

    def _sync(self, flow: JobFlow) -> None:
        if not self._validate(flow):
            if flow.status.get("state") != "Failed":
                flow.status["state"] = "Failed"
                flow.status["reason"] = "cycle in dependsOn graph"
                self.store.update("JobFlow", flow)
            return

        states = {}
        for step in flow.flows:
            job = self.store.get("Job", flow.meta.namespace,
                                 self._job_name(flow, step.name))
            states[step.name] = job.status.phase if job else None

        changed = False
        for step in flow.flows:
            if states[step.name] is not None:
                continue
            deps_ok = all(states.get(d) == JobPhase.COMPLETED.value
                          for d in step.depends_on)
            if not deps_ok:
                continue
            tmpl = self.store.get("JobTemplate", flow.meta.namespace, step.name)
            if tmpl is None:
                continue
            spec = copy.deepcopy(tmpl.spec)
            for k, v in (step.patch or {}).items():
                setattr(spec, k, v)
            self.store.create("Job", Job(
                meta=ObjectMeta(name=self._job_name(flow, step.name),
                                namespace=flow.meta.namespace,
                                labels={"flow.volcano.sh/jobtemplate":
                                        step.name},
                                owner=f"JobFlow/{flow.meta.key}"),
                spec=spec))
            states[step.name] = JobPhase.PENDING.value
            changed = True

        if all(v == JobPhase.COMPLETED.value for v in states.values()) \
                and states:
            if flow.status.get("state") != "Succeeded":
                flow.status["state"] = "Succeeded"
                self.store.update("JobFlow", flow)
                if flow.job_retain_policy == "delete":
                    for step in flow.flows:
                        self.store.delete("Job", flow.meta.namespace,
                                          self._job_name(flow, step.name))
        else:
            running = "Running" if any(v is not None for v in states.values()) \
                else "Pending"
            if flow.status.get("state") != running:
                flow.status["state"] = running
                self.store.update("JobFlow", flow)
            elif changed:
                self.store.update("JobFlow", flow)
