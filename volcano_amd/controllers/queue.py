"""Queue controller (reference ``pkg/controllers/queue/``): maintains
queue status (Open/Closed/Closing + podgroup counts) and reacts to bus
Commands (OpenQueue/CloseQueue)."""

from __future__ import annotations

from ..api.objects import ObjectMeta, Queue
from ..api.types import Action, PodGroupPhase, QueueState
from ..store import EventType
from .framework import Controller, register


@register("queue")
class QueueController(Controller):
    watch_kinds = ("Queue", "PodGroup", "Command")

    def initialize(self, store) -> None:
        super().initialize(store)
        self._dirty = set()
        if store.get("Queue", "default", "default") is None:
            store.create("Queue", Queue(meta=ObjectMeta(name="default")))

    def handle(self, ev) -> None:
        if ev.kind == "Queue" and ev.type != EventType.DELETED:
            self._dirty.add(ev.obj.meta.name)
        elif ev.kind == "PodGroup":
            self._dirty.add(ev.obj.spec.queue)
        elif ev.kind == "Command" and ev.type != EventType.DELETED:
            from ..utils.features import enabled
            if not enabled("QueueCommandSync"):
                return
            cmd = ev.obj
            if cmd.target_kind != "Queue":
                return
            q = self.store.get("Queue", "default", cmd.target_name)
            if q is not None:
                if cmd.action == Action.CLOSE_QUEUE.value:
                    # Closing while podgroups remain; Closed once drained
                    # (reference queue/state machine factory.go:39-52)
                    active = self.store.list(
                        "PodGroup",
                        selector=lambda g: g.spec.queue == q.meta.name)
                    q.status.state = QueueState.CLOSING.value if active \
                        else QueueState.CLOSED.value
                elif cmd.action == Action.OPEN_QUEUE.value:
                    q.status.state = QueueState.OPEN.value
                self.store.update("Queue", q)
            self.store.delete("Command", cmd.meta.namespace, cmd.meta.name)

    def resync(self) -> None:
        dirty, self._dirty = self._dirty, set()
        for name in dirty:
            q = self.store.get("Queue", "default", name)
            if q is None:
                continue
            pgs = self.store.list("PodGroup",
                                  selector=lambda g: g.spec.queue == name)
            old = (q.status.pending, q.status.running, q.status.inqueue)
            q.status.pending = sum(
                1 for g in pgs if g.status.phase == PodGroupPhase.PENDING.value)
            q.status.running = sum(
                1 for g in pgs if g.status.phase == PodGroupPhase.RUNNING.value)
            q.status.inqueue = sum(
                1 for g in pgs if g.status.phase == PodGroupPhase.INQUEUE.value)
            changed = (q.status.pending, q.status.running,
                       q.status.inqueue) != old
            if q.status.state == QueueState.CLOSING.value and not pgs:
                q.status.state = QueueState.CLOSED.value
                changed = True
            if changed:
                self.store.update("Queue", q)
