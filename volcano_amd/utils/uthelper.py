"""Declarative scheduler-in-a-box test harness.

Port of the reference's ``pkg/scheduler/uthelper/helper.go``
(TestCommonStruct: declare Pods/Nodes/PodGroups/Queues + plugins, Run
actions against a mock cache with fake binder/evictor, CheckAll asserts
ExpectBindMap/ExpectEvicted/ExpectStatus).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List, Optional

from ..api.objects import HyperNode, Node, Pod, PodGroup, Queue
from ..scheduler import FakeBinder, Scheduler, SchedulerCache
from ..scheduler.config import PluginOption, Tier, default_config
from ..store import ObjectStore


@dataclass
class TestCommonStruct:
    __test__ = False          # not a pytest class (harness, like uthelper)

    name: str = ""
    nodes: List[Node] = field(default_factory=list)
    pods: List[Pod] = field(default_factory=list)
    podgroups: List[PodGroup] = field(default_factory=list)
    queues: List[Queue] = field(default_factory=list)
    hypernodes: List[HyperNode] = field(default_factory=list)
    #: plugin tiers by name (None → reference default tiers)
    tiers: Optional[List[List[str]]] = None
    plugin_args: Dict[str, dict] = field(default_factory=dict)
    actions: Optional[List[str]] = None
    cycles: int = 1

    # expectations (CheckAll)
    expect_bind_map: Optional[Dict[str, str]] = None   # task key → node
    expect_bind_count: Optional[int] = None
    expect_evicted: Optional[List[str]] = None
    expect_pipelined: Optional[int] = None
    expect_status: Optional[Dict[str, str]] = None     # podgroup key → phase

    def run(self) -> "TestCommonStruct":
        self.store = ObjectStore()
        for n in self.nodes:
            self.store.create("Node", n)
        for q in self.queues:
            self.store.create("Queue", q)
        for pg in self.podgroups:
            self.store.create("PodGroup", pg)
        for p in self.pods:
            self.store.create("Pod", p)
        for h in self.hypernodes:
            self.store.create("HyperNode", h)

        config = default_config()
        if self.tiers is not None:
            config.tiers = [
                Tier(plugins=[PluginOption(n, arguments=self.plugin_args.get(n, {}))
                              for n in tier])
                for tier in self.tiers]
        if self.actions is not None:
            config.actions = self.actions
        self.binder = FakeBinder()
        self.cache = SchedulerCache(store=self.store, binder=self.binder)
        self.scheduler = Scheduler(self.cache, config)
        for _ in range(self.cycles):
            self.scheduler.run_once()
        return self

    def check_all(self) -> None:
        ctx = f"[{self.name}] " if self.name else ""
        if self.expect_bind_count is not None:
            assert len(self.binder.binds) == self.expect_bind_count, \
                f"{ctx}bind count {len(self.binder.binds)} != " \
                f"{self.expect_bind_count}: {self.binder.binds}"
        if self.expect_bind_map is not None:
            assert self.binder.binds == self.expect_bind_map, \
                f"{ctx}bind map mismatch: {self.binder.binds}"
        if self.expect_evicted is not None:
            assert sorted(self.binder.evictions) == sorted(self.expect_evicted), \
                f"{ctx}evictions {self.binder.evictions}"
        if self.expect_pipelined is not None:
            from ..api.types import TaskStatus
            got = sum(len(j.task_status_index.get(TaskStatus.PIPELINED, ()))
                      for j in self.cache.jobs.values())
            assert got == self.expect_pipelined, \
                f"{ctx}pipelined {got} != {self.expect_pipelined}"
        if self.expect_status is not None:
            for key, phase in self.expect_status.items():
                job = self.cache.jobs.get(key)
                assert job is not None and job.phase == phase, \
                    f"{ctx}{key} phase {job.phase if job else None} != {phase}"
