"""In-process object store — the framework's kube-apiserver/etcd stand-in.

The reference's only durable state and inter-component bus is the
kube-apiserver (SURVEY.md §5 "checkpoint/resume": everything is watch/
list + update against etcd).  This environment has no cluster, so the
control plane runs against this store: versioned typed objects, list/get/
create/update/delete, and watch channels with replay from a resource
version — the same contract client-go informers give the reference's
cache (`pkg/scheduler/cache/cache.go:636-794`).

Thread-safe; watches are bounded queues drained by consumer threads
(controllers, the scheduler cache).  A JSON snapshot/restore pair stands
in for etcd durability (reference: state is rebuilt from informers at
startup — here ``save``/``load`` give the same crash-resume story).
"""

from __future__ import annotations

import enum
import json
import queue
import threading
import uuid
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

from ..api.objects import KINDS, from_dict, to_dict


class EventType(str, enum.Enum):
    ADDED = "ADDED"
    MODIFIED = "MODIFIED"
    DELETED = "DELETED"


@dataclass
class Event:
    type: EventType
    kind: str
    obj: object
    resource_version: int


class Watch:
    """A bounded event stream for one consumer (informer analog)."""

    def __init__(self, store: "ObjectStore", kinds: Tuple[str, ...],
                 maxsize: int = 100000):
        self._store = store
        self.kinds = kinds
        self._q: "queue.Queue[Optional[Event]]" = queue.Queue(maxsize=maxsize)
        self._closed = False
        # backpressure: a slow consumer must never block writers — when
        # the queue is full the event is DROPPED and the stream marked
        # overflowed; the consumer detects it and relists from the store
        # (k8s "watch too old" → informer resync semantics)
        self.overflowed = False

    def _push(self, ev: Event) -> None:
        if not self._closed and (not self.kinds or ev.kind in self.kinds):
            try:
                self._q.put_nowait(ev)
            except queue.Full:
                self.overflowed = True

    def next(self, timeout: Optional[float] = None) -> Optional[Event]:
        try:
            return self._q.get(timeout=timeout)
        except queue.Empty:
            return None

    def drain(self) -> List[Event]:
        out: List[Event] = []
        while True:
            try:
                ev = self._q.get_nowait()
            except queue.Empty:
                return out
            if ev is not None:
                out.append(ev)

    def stop(self) -> None:
        self._closed = True
        self._store._remove_watch(self)
        try:
            self._q.put_nowait(None)    # wake a blocked next(); best-effort
        except queue.Full:
            pass                        # consumer has pending events anyway


class ObjectStore:
    """Versioned, watchable object store keyed by (kind, namespace/name)."""

    def __init__(self, journal_size: int = 200000):
        self._lock = threading.RLock()
        self._objects: Dict[str, Dict[str, object]] = {k: {} for k in KINDS}
        self._rv = 0
        self._watches: List[Watch] = []
        # event journal for HTTP watch replay: (rv, type, kind, obj_snapshot)
        from collections import deque
        self._journal = deque(maxlen=journal_size)

    # -- internals -----------------------------------------------------------
    def _bump(self) -> int:
        self._rv += 1
        return self._rv

    def _notify(self, ev: Event) -> None:
        for w in list(self._watches):
            w._push(ev)
        self._journal.append((ev.resource_version, ev.type.value, ev.kind,
                              to_dict(ev.obj)))

    def journal_since(self, rv: int, kinds: Optional[Tuple[str, ...]] = None):
        """Events with resource_version > rv (HTTP watch long-poll)."""
        with self._lock:
            return [(v, t, k, o) for (v, t, k, o) in self._journal
                    if v > rv and (not kinds or k in kinds)]

    def _remove_watch(self, w: Watch) -> None:
        with self._lock:
            if w in self._watches:
                self._watches.remove(w)

    @staticmethod
    def _key(obj) -> str:
        return obj.meta.key

    # -- CRUD ----------------------------------------------------------------
    def create(self, kind: str, obj) -> object:
        with self._lock:
            key = self._key(obj)
            if key in self._objects[kind]:
                raise KeyError(f"{kind} {key} already exists")
            if not obj.meta.uid:
                obj.meta.uid = uuid.uuid4().hex
            obj.meta.resource_version = self._bump()
            self._objects[kind][key] = obj
            ev = Event(EventType.ADDED, kind, obj, obj.meta.resource_version)
            self._notify(ev)
            return obj

    def update(self, kind: str, obj) -> object:
        with self._lock:
            key = self._key(obj)
            if key not in self._objects[kind]:
                raise KeyError(f"{kind} {key} not found")
            obj.meta.resource_version = self._bump()
            self._objects[kind][key] = obj
            self._notify(Event(EventType.MODIFIED, kind, obj,
                               obj.meta.resource_version))
            return obj

    def apply(self, kind: str, obj) -> object:
        """Create-or-update (kubectl apply semantics)."""
        with self._lock:
            if self._key(obj) in self._objects[kind]:
                return self.update(kind, obj)
            return self.create(kind, obj)

    def delete(self, kind: str, namespace: str, name: str) -> Optional[object]:
        with self._lock:
            key = f"{namespace}/{name}"
            obj = self._objects[kind].pop(key, None)
            if obj is not None:
                self._notify(Event(EventType.DELETED, kind, obj, self._bump()))
            return obj

    def get(self, kind: str, namespace: str, name: str) -> Optional[object]:
        with self._lock:
            return self._objects[kind].get(f"{namespace}/{name}")

    def list(self, kind: str, namespace: Optional[str] = None,
             selector: Optional[Callable[[object], bool]] = None) -> List[object]:
        with self._lock:
            out = list(self._objects[kind].values())
        if namespace is not None:
            out = [o for o in out if o.meta.namespace == namespace]
        if selector is not None:
            out = [o for o in out if selector(o)]
        return out

    def count(self, kind: str) -> int:
        with self._lock:
            return len(self._objects[kind])

    @property
    def resource_version(self) -> int:
        with self._lock:
            return self._rv

    # -- watch ---------------------------------------------------------------
    def watch(self, *kinds: str, replay: bool = True,
              maxsize: int = 1000000) -> Watch:
        """Open an event stream; with ``replay`` the current state arrives
        first as ADDED events (informer initial-list semantics,
        reference pkg/schedulercommon/cache)."""
        with self._lock:
            w = Watch(self, kinds, maxsize=maxsize)
            if replay:
                for kind in (kinds or tuple(self._objects)):
                    for obj in self._objects[kind].values():
                        w._push(Event(EventType.ADDED, kind, obj,
                                      obj.meta.resource_version))
            self._watches.append(w)
            return w

    # -- durability (etcd stand-in) -------------------------------------------
    def save(self, path: str) -> None:
        with self._lock:
            data = {
                "rv": self._rv,
                "objects": {
                    kind: {k: to_dict(o) for k, o in objs.items()}
                    for kind, objs in self._objects.items()
                },
            }
        with open(path, "w") as f:
            json.dump(data, f)

    @classmethod
    def load(cls, path: str) -> "ObjectStore":
        with open(path) as f:
            data = json.load(f)
        store = cls()
        store._rv = int(data.get("rv", 0))
        for kind, objs in data.get("objects", {}).items():
            cls_t = KINDS.get(kind)
            if cls_t is None:
                continue
            for key, od in objs.items():
                store._objects[kind][key] = from_dict(cls_t, od)
        return store
