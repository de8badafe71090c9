"""Statement — transactional eviction/pipeline ops with rollback.

Reference: ``pkg/scheduler/framework/statement.go`` (Evict/Pipeline/
Allocate buffered against session state, Commit to cache or reverse-order
Discard — the gang primitive).

Division of labor in the MI355X design: *allocation* transactions run on
device (the select/finalize/revert kernels ARE the statement for the
allocate path — scheduler_kernels.hip); this host-side Statement covers
the preempt/reclaim paths, where evictions mutate the host info mirror
and must roll back if the preemptor gang cannot reach its minimum.
"""

from __future__ import annotations

from typing import List, Tuple

from ..api.info import TaskInfo
from ..api.types import TaskStatus


class Statement:
    def __init__(self, ssn):
        self.ssn = ssn
        self.ops: List[Tuple] = []

    # -- operations (applied to session state immediately) -------------------
    def evict(self, victim: TaskInfo, reason: str = "preempted") -> None:
        job = self.ssn.jobs.get(victim.job_key)
        node = self.ssn.nodes.get(victim.node_name)
        prev = victim.status
        if node is not None:
            node.remove_task(victim)
        if job is not None:
            job.update_task_status(victim, TaskStatus.RELEASING)
        else:
            victim.status = TaskStatus.RELEASING
        if node is not None:
            node.add_task(victim)       # re-adds under releasing accounting
        self.ops.append(("evict", victim, prev, reason))

    def pipeline(self, task: TaskInfo, node_name: str) -> None:
        job = self.ssn.jobs.get(task.job_key)
        node = self.ssn.nodes.get(node_name)
        task.node_name = node_name
        if job is not None:
            job.update_task_status(task, TaskStatus.PIPELINED)
        else:
            task.status = TaskStatus.PIPELINED
        if node is not None:
            node.add_task(task)
        self.ops.append(("pipeline", task, node_name))

    # -- outcome --------------------------------------------------------------
    def discard(self) -> None:
        """Reverse-order undo (statement.go:375)."""
        for op in reversed(self.ops):
            if op[0] == "evict":
                _, victim, prev, _ = op
                node = self.ssn.nodes.get(victim.node_name)
                job = self.ssn.jobs.get(victim.job_key)
                if node is not None:
                    node.remove_task(victim)
                if job is not None:
                    job.update_task_status(victim, prev)
                else:
                    victim.status = prev
                if node is not None:
                    node.add_task(victim)
            elif op[0] == "pipeline":
                _, task, node_name = op
                node = self.ssn.nodes.get(node_name)
                job = self.ssn.jobs.get(task.job_key)
                if node is not None:
                    node.remove_task(task)
                task.node_name = ""
                if job is not None:
                    job.update_task_status(task, TaskStatus.PENDING)
                else:
                    task.status = TaskStatus.PENDING
        self.ops.clear()

    def commit(self) -> None:
        """Flush evictions to the cache/binder (statement.go:402); pipelined
        tasks hold their reservation for this cycle (see engine notes)."""
        cache = self.ssn.cache
        for op in self.ops:
            if op[0] == "evict":
                _, victim, _, reason = op
                cache.binder.evict(victim, reason)
        self.ops.clear()
