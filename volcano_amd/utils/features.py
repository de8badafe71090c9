"""Feature gates (reference ``pkg/features/volcano_features.go`` — the
k8s component-base featuregate wrapped around volcano's optional
behaviors).  All gates default ON, matching the reference's defaults;
they flip via scheduler conf YAML (``feature_gates: {Name: false}``),
``set_gates``, or the ``VAMD_FEATURE_GATES`` env
(``"A=false,B=true"``).

Gate → behavior map here:

* ``WorkLoadSupport`` — the podgroup controller wraps plain pods into
  implicit PodGroups.
* ``VolcanoJobSupport`` — the job controller materializes vcjobs.
* ``PodDisruptionBudgetsSupport`` — the pdb plugin filters victims.
* ``QueueCommandSync`` — the queue controller consumes Command objects.
* ``PriorityClass`` — priority plugin ordering/preemption.
* ``CSIStorage`` — volume zone filter + bind-time claim matching.
* ``ResourceTopology`` — numaaware filtering.
* ``CronVolcanoJobSupport`` — the cronjob controller.
* ``SchedulingGatesQueueAdmission`` — enqueue lifts the
  queue-allocation scheduling gate on admission.
"""

from __future__ import annotations

import os
from typing import Dict

DEFAULT_GATES = {
    "WorkLoadSupport": True,
    "VolcanoJobSupport": True,
    "PodDisruptionBudgetsSupport": True,
    "QueueCommandSync": True,
    "PriorityClass": True,
    "CSIStorage": True,
    "ResourceTopology": True,
    "CronVolcanoJobSupport": True,
    "SchedulingGatesQueueAdmission": True,
}

_gates: Dict[str, bool] = dict(DEFAULT_GATES)


def _load_env() -> None:
    raw = os.environ.get("VAMD_FEATURE_GATES", "")
    for part in raw.split(","):
        if "=" in part:
            k, v = part.split("=", 1)
            k = k.strip()
            if k:
                _gates[k] = v.strip().lower() in ("1", "true", "yes", "on")


_load_env()


def enabled(name: str) -> bool:
    """Unknown gates are ON (forward-compatible, like the reference)."""
    return _gates.get(name, True)


def set_gates(gates: Dict[str, bool]) -> None:
    for k, v in (gates or {}).items():
        _gates[k] = bool(v)


def reset() -> None:
    _gates.clear()
    _gates.update(DEFAULT_GATES)
    _load_env()
