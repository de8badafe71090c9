"""Core enums: task status, pod-group phases, lifecycle events/actions.

Mirrors the reference's ``pkg/scheduler/api/types.go`` (TaskStatus bitmask
state machine), ``apis/scheduling/v1beta1/types.go`` (PodGroup phases) and
``apis/bus/v1alpha1/{events,actions}.go`` (lifecycle policy vocabulary).
"""

from __future__ import annotations

import enum


class TaskStatus(enum.IntEnum):
    """Status of a task (pod) inside the scheduler.

    Reference: api/types.go:33-66 — Pending → Allocated → Pipelined →
    Binding → Bound → Running → Releasing → Succeeded/Failed.
    """

    PENDING = 0      # waiting in queue
    ALLOCATED = 1    # resources assigned this cycle (on idle)
    PIPELINED = 2    # assigned on *future* idle (waiting for a release)
    BINDING = 3      # handed to the binder
    BOUND = 4        # bind confirmed
    RUNNING = 5
    RELEASING = 6    # being evicted / terminating
    SUCCEEDED = 7
    FAILED = 8
    UNKNOWN = 9

    @property
    def occupies_node(self) -> bool:
        """Does this status consume node resources now? (api/types.go
        AllocatedStatus)."""
        return self in (TaskStatus.ALLOCATED, TaskStatus.BINDING,
                        TaskStatus.BOUND, TaskStatus.RUNNING)


ALLOCATED_STATUSES = (TaskStatus.ALLOCATED, TaskStatus.BINDING,
                      TaskStatus.BOUND, TaskStatus.RUNNING)


class PodGroupPhase(str, enum.Enum):
    """scheduling/v1beta1 PodGroup phases (types.go:174-240)."""

    PENDING = "Pending"
    INQUEUE = "Inqueue"
    RUNNING = "Running"
    UNKNOWN = "Unknown"
    COMPLETED = "Completed"


class JobPhase(str, enum.Enum):
    """batch/v1alpha1 Job phases (job controller state machine states,
    pkg/controllers/job/state/factory.go)."""

    PENDING = "Pending"
    ABORTING = "Aborting"
    ABORTED = "Aborted"
    RUNNING = "Running"
    RESTARTING = "Restarting"
    COMPLETING = "Completing"
    COMPLETED = "Completed"
    TERMINATING = "Terminating"
    TERMINATED = "Terminated"
    FAILED = "Failed"


class QueueState(str, enum.Enum):
    """Queue lifecycle (queue controller state machine)."""

    OPEN = "Open"
    CLOSED = "Closed"
    CLOSING = "Closing"
    UNKNOWN = "Unknown"


class Event(str, enum.Enum):
    """Lifecycle events (bus/v1alpha1/events.go)."""

    ANY = "*"
    POD_FAILED = "PodFailed"
    POD_EVICTED = "PodEvicted"
    POD_PENDING = "PodPending"
    POD_RUNNING = "PodRunning"
    UNKNOWN = "Unknown"
    TASK_COMPLETED = "TaskCompleted"
    TASK_FAILED = "TaskFailed"
    OUT_OF_SYNC = "OutOfSync"
    COMMAND_ISSUED = "CommandIssued"
    JOB_UPDATED = "JobUpdated"


class Action(str, enum.Enum):
    """Lifecycle actions (bus/v1alpha1/actions.go:19-39)."""

    ABORT_JOB = "AbortJob"
    RESTART_JOB = "RestartJob"
    RESTART_TASK = "RestartTask"
    RESTART_POD = "RestartPod"
    TERMINATE_JOB = "TerminateJob"
    COMPLETE_JOB = "CompleteJob"
    RESUME_JOB = "ResumeJob"
    SYNC_JOB = "SyncJob"
    ENQUEUE_JOB = "EnqueueJob"
    SYNC_QUEUE = "SyncQueue"
    OPEN_QUEUE = "OpenQueue"
    CLOSE_QUEUE = "CloseQueue"
