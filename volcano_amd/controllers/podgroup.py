"""PodGroup controller (reference ``pkg/controllers/podgroup/``):
auto-creates a minMember=1 PodGroup for *normal* pods (pods not owned by
a vcjob and without a group annotation) so they pass the gang machinery
(pg_controller_handler.go:301 createNormalPodPGIfNotExist)."""

from __future__ import annotations

from ..api.objects import (ANN_PODGROUP, ANN_QUEUE, LBL_JOB_NAME, ObjectMeta,
                           PodGroup, PodGroupSpec)
from ..store import EventType
from .framework import Controller, register


@register("podgroup")
class PodGroupController(Controller):
    watch_kinds = ("Pod",)

    def handle(self, ev) -> None:
        from ..utils.features import enabled
        if not enabled("WorkLoadSupport"):
            return          # plain-pod wrapping is the workload gate
        if ev.type == EventType.DELETED:
            return
        pod = ev.obj
        if pod.meta.labels.get(LBL_JOB_NAME):
            return                      # vcjob pods: job controller owns the group
        if pod.meta.annotations.get(ANN_PODGROUP):
            return
        pg_name = f"podgroup-{pod.meta.uid or pod.meta.name}"
        if self.store.get("PodGroup", pod.meta.namespace, pg_name) is None:
            self.store.create("PodGroup", PodGroup(
                meta=ObjectMeta(name=pg_name, namespace=pod.meta.namespace,
                                owner=f"Pod/{pod.meta.key}"),
                spec=PodGroupSpec(
                    min_member=1,
                    queue=pod.meta.annotations.get(ANN_QUEUE, "default"),
                    min_resources=pod.request.clone())))
        pod.meta.annotations[ANN_PODGROUP] = pg_name
        self.store.update("Pod", pod)
