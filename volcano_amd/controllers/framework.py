"""Controller framework (reference ``pkg/controllers/framework/
interface.go:51-64``: Name/Initialize/Run + registry; controllers are
registered by side-effect import in cmd/controller-manager/main.go).

Controllers here are event-driven over store watches with an explicit
``sync_once`` (drain + handle) so tests and the manager loop share one
deterministic code path.
"""

from __future__ import annotations

import threading
import time
from typing import Dict, List, Optional, Type

from ..store import ObjectStore


class Controller:
    name = "controller"
    watch_kinds: tuple = ()

    def initialize(self, store: ObjectStore) -> None:
        self.store = store
        self._watch = store.watch(*self.watch_kinds) if self.watch_kinds \
            else None

    def handle(self, ev) -> None:  # pragma: no cover - interface
        pass

    def resync(self) -> None:
        """Periodic full pass (cron triggers, TTL sweeps)."""

    def sync_once(self) -> int:
        from ..utils.metrics import METRICS
        n = 0
        if self._watch is not None:
            for ev in self._watch.drain():
                self.handle(ev)
                n += 1
        self.resync()
        if n:
            # reference controllers/metrics: per-controller event counters
            METRICS.inc(f"controller_handled_events:{self.name}", n)
        return n


CONTROLLER_REGISTRY: Dict[str, Type[Controller]] = {}


def register(name: str):
    def deco(cls):
        cls.name = name
        CONTROLLER_REGISTRY[name] = cls
        return cls
    return deco


class ControllerManager:
    """controller-manager main loop: owns all registered controllers."""

    def __init__(self, store: ObjectStore,
                 controllers: Optional[List[str]] = None,
                 period: float = 0.05):
        self.store = store
        self.period = period
        names = controllers if controllers is not None \
            else list(CONTROLLER_REGISTRY)
        self.controllers: List[Controller] = []
        for n in names:
            c = CONTROLLER_REGISTRY[n]()
            c.initialize(store)
            self.controllers.append(c)
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None

    def sync_once(self) -> int:
        return sum(c.sync_once() for c in self.controllers)

    def sync_until_quiet(self, max_rounds: int = 50) -> None:
        """Drive controllers to a fixed point (test helper)."""
        for _ in range(max_rounds):
            if self.sync_once() == 0:
                return

    def run(self) -> None:
        def loop():
            while not self._stop.is_set():
                self.sync_once()
                self._stop.wait(self.period)
        self._thread = threading.Thread(target=loop, daemon=True)
        self._thread.start()

    def stop(self) -> None:
        self._stop.set()
        if self._thread:
            self._thread.join(timeout=2.0)
