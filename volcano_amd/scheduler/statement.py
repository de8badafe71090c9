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
        note = getattr(self.ssn.cache, "note_pipelined", None)
        if note is not None:
            note(task.job_key)
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

    # -- composition / replay (statement.go:433 Merge, :440 SaveOperations,
    #    :457 RecoverOperations) ----------------------------------------------
    def merge(self, *stmts: "Statement") -> None:
        """Absorb other statements' pending ops so they commit or discard
        with this one; sources are cleared (no double-commit)."""
        for st in stmts:
            self.ops.extend(st.ops)
            st.ops.clear()

    def save_operations(self) -> List[Tuple]:
        """Detached record of the pending ops (task identity + target),
        usable to replay the same decisions in a later session via
        :meth:`recover_operations` — the cross-action replay primitive."""
        saved = []
        for op in self.ops:
            if op[0] == "evict":
                _, victim, _, reason = op
                saved.append(("evict", victim.key, reason))
            elif op[0] == "pipeline":
                _, task, node_name = op
                saved.append(("pipeline", task.key, node_name))
        return saved

    def recover_operations(self, saved: List[Tuple]) -> bool:
        """Re-apply a saved op list against THIS session's live state.
        Tasks are looked up by key (a later session rebuilds its infos);
        missing tasks abort the replay and roll back (all-or-nothing,
        mirroring the reference's error-out)."""
        by_key = {}
        for job in self.ssn.jobs.values():
            for t in job.tasks.values():
                by_key[t.key] = t
        start = len(self.ops)
        for op in saved:
            t = by_key.get(op[1])
            if t is None:
                # roll back only what this replay added
                tail = Statement(self.ssn)
                tail.ops = self.ops[start:]
                tail.discard()
                del self.ops[start:]
                return False
            if op[0] == "evict":
                self.evict(t, op[2])
            elif op[0] == "pipeline":
                self.pipeline(t, op[2])
        return True

    def has_evictions(self) -> bool:
        return any(op[0] == "evict" for op in self.ops)

    def commit(self) -> None:
        """Flush evictions to the cache/binder (statement.go:402); pipelined
        tasks hold their reservation for this cycle (see engine notes)."""
        cache = self.ssn.cache
        for op in self.ops:
            if op[0] == "evict":
                _, victim, _, reason = op
                cache.binder.evict(victim, reason)
                # release hooks (deviceshare/numaaware pools) — the
                # counterpart of fire_allocate (reference statement.go:402
                # fires EventHandler.DeallocateFunc on commit)
                self.ssn.fire_evict(victim)
        self.ops.clear()
