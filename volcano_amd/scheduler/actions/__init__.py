"""Per-cycle algorithm pipeline (reference ``pkg/scheduler/actions/``).

Registry mirrors ``actions/factory.go``.
"""

from .allocate import AllocateAction
from .backfill import BackfillAction
from .enqueue import EnqueueAction
from .gangpreempt import GangPreemptAction, GangReclaimAction
from .preempt import PreemptAction
from .reclaim import ReclaimAction
from .shuffle import ShuffleAction

ACTION_REGISTRY = {
    "enqueue": EnqueueAction,
    "allocate": AllocateAction,
    "backfill": BackfillAction,
    "preempt": PreemptAction,
    "reclaim": ReclaimAction,
    "gangpreempt": GangPreemptAction,
    "gangreclaim": GangReclaimAction,
    "shuffle": ShuffleAction,
}


def new_action(name: str):
    try:
        return ACTION_REGISTRY[name]()
    except KeyError:
        raise KeyError(f"unknown action {name!r}; known: {sorted(ACTION_REGISTRY)}")
