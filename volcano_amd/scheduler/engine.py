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
        if self.config.feature_gates:
            from ..utils import features
            features.set_gates(self.config.feature_gates)
        self._stop = threading.Event()
        self._actions = [actions_mod.new_action(a) for a in self.config.actions]
        # real-usage metrics source (reference cache.setMetricsData,
        # cache.go:1808): periodically publishes node usage annotations
        # the usage plugin / oversubscription consume
        self._metrics_source = None
        self._metrics_interval = 30.0
        self._metrics_last = 0.0
        m = dict(self.config.metrics or {})
        if m.get("type") and m.get("type") != "annotation":
            from .metrics_source import new_source
            self._metrics_interval = float(m.pop("interval", 30.0))
            kind = m.pop("type")
            try:
                self._metrics_source = new_source(kind, **m)
            except Exception:
                self._metrics_source = None

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
        # plugin closures reference the session and live in its registries
        # — drop them so sessions are reclaimed by refcount alone (no
        # reliance on the cyclic GC in the long-running daemon)
        for name in ("job_order_fns", "queue_order_fns", "task_order_fns",
                     "job_order_keys", "queue_order_keys", "job_valid_fns",
                     "job_ready_fns", "job_pipelined_fns",
                     "job_enqueueable_fns", "job_enqueueable_bulk_fns",
                     "job_starving_fns",
                     "overused_fns", "allocatable_fns", "preemptable_fns",
                     "reclaimable_fns", "victim_tasks_fns",
                     "victim_filter_fns", "event_handlers",
                     "class_constraint_hooks", "class_bias_fns", "plugins"):
            getattr(ssn, name, []).clear()
        if hasattr(ssn, "job_enqueued_fns"):
            ssn.job_enqueued_fns.clear()

    def _sync_metrics(self) -> None:
        src = self._metrics_source
        if src is None or self.cache.store is None:
            return
        now = time.monotonic()
        if now - self._metrics_last < self._metrics_interval:
            return
        self._metrics_last = now
        from .metrics_source import ANN_CPU_USAGE, ANN_MEM_USAGE
        nodes = self.cache.store.list("Node")
        usage = src.node_usage(nodes)
        for node in nodes:
            u = usage.get(node.meta.name)
            if not u:
                continue
            ann = node.meta.annotations
            new = (str(u["cpu"]), str(u["memory"]))
            if (ann.get(ANN_CPU_USAGE), ann.get(ANN_MEM_USAGE)) != new:
                ann[ANN_CPU_USAGE], ann[ANN_MEM_USAGE] = new
                self.cache.store.update("Node", node)

    # -- cycle ---------------------------------------------------------------
    def run_once(self) -> Session:
        t0 = time.perf_counter()
        self._sync_metrics()
        ssn = self.open_session()
        timings = {}
        for action in self._actions:
            ta = time.perf_counter()
            action.execute(ssn)
            dt = time.perf_counter() - ta
            timings[action.name] = dt
            METRICS.observe(f"action_scheduling_latency:{action.name}", dt)
        self.close_session(ssn)
        METRICS.observe("e2e_scheduling_latency", time.perf_counter() - t0)
        self._record_cycle(ssn, t0, timings)
        return ssn

    _cycle_idx = 0

    def _record_cycle(self, ssn, t0, timings) -> None:
        """Per-cycle decision recorder (reference actions/allocate
        recorder.go): one JSON line per cycle to $VAMD_CYCLE_LOG —
        enough to reconstruct what the cycle saw and decided."""
        import os
        path = os.environ.get("VAMD_CYCLE_LOG")
        if not path:
            return
        import json
        from ..api.types import TaskStatus
        pending = sum(len(j.task_status_index.get(TaskStatus.PENDING, ()))
                      for j in ssn.jobs.values())
        self._cycle_idx += 1
        rec = {
            "cycle": self._cycle_idx,
            "ts": time.time(),
            "e2e_ms": round((time.perf_counter() - t0) * 1000, 3),
            "actions_ms": {k: round(v * 1000, 3) for k, v in timings.items()},
            "nodes": len(ssn.nodes),
            "jobs": len(ssn.jobs),
            "queues": len(ssn.queues),
            "pending_after": pending,
        }
        try:
            with open(path, "a") as f:
                f.write(json.dumps(rec) + "\n")
        except OSError:
            pass

    def _idle(self) -> bool:
        """Nothing to do: no store events since last cycle and no pending
        or pipelined work anywhere (daemon fast path; the reference runs
        a full snapshot every second regardless)."""
        if self.cache.sync() > 0:
            return False
        from ..api.types import TaskStatus
        for job in self.cache.jobs.values():
            idx = job.task_status_index
            if idx.get(TaskStatus.PENDING) or idx.get(TaskStatus.PIPELINED):
                return False
        return True

    def run(self, period: Optional[float] = None) -> None:
        period = period if period is not None else self.config.schedule_period
        while not self._stop.is_set():
            start = time.perf_counter()
            if not self._idle():
                self.run_once()
            elapsed = time.perf_counter() - start
            self._stop.wait(max(0.0, period - elapsed))

    def stop(self) -> None:
        self._stop.set()
