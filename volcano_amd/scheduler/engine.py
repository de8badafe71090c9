"""Scheduler daemon: the per-cycle loop (reference
``pkg/scheduler/scheduler.go`` Run/runOnce + ``framework/framework.go``
OpenSession/CloseSession)."""

from __future__ import annotations

import threading
import time
from typing import Optional

from ..utils.metrics import METRICS
from . import actions as actions_mod
from . import plugins as plugins_mod
from .cache import SchedulerCache
from .config import SchedulerConfiguration, default_config
from .session import Session


class Scheduler:
    def __init__(self, cache: SchedulerCache,
                 config: Optional[SchedulerConfiguration] = None):
        self.cache = cache
        self.config = config or default_config()
        self._stop = threading.Event()
        self._actions = [actions_mod.new_action(a) for a in self.config.actions]

    # -- session lifecycle (framework.go:34-71) ------------------------------
    def open_session(self) -> Session:
        t0 = time.perf_counter()
        ssn = Session(self.cache, self.config)
        self.cache.snapshot_into(ssn)
        ssn.build_queue_tensors()
        for tier in self.config.tiers:
            ssn.open_tier()
            for opt in tier.plugins:
                plugin = plugins_mod.new_plugin(opt.name, opt.arguments)
                ssn.plugins.append(plugin)
                plugin.on_session_open(ssn)
        METRICS.observe("open_session_duration", time.perf_counter() - t0)
        return ssn

    def close_session(self, ssn: Session) -> None:
        for plugin in ssn.plugins:
            plugin.on_session_close(ssn)

    # -- cycle ---------------------------------------------------------------
    def run_once(self) -> Session:
        t0 = time.perf_counter()
        ssn = self.open_session()
        for action in self._actions:
            ta = time.perf_counter()
            action.execute(ssn)
            METRICS.observe(f"action_scheduling_latency:{action.name}",
                            time.perf_counter() - ta)
        self.close_session(ssn)
        METRICS.observe("e2e_scheduling_latency", time.perf_counter() - t0)
        return ssn

    def run(self, period: Optional[float] = None) -> None:
        period = period if period is not None else self.config.schedule_period
        while not self._stop.is_set():
            start = time.perf_counter()
            self.run_once()
            elapsed = time.perf_counter() - start
            self._stop.wait(max(0.0, period - elapsed))

    def stop(self) -> None:
        self._stop.set()
