"""FakeKubelet — drives bound pods through Running/Succeeded/Failed.

The test-tier analog of KWOK's virtual kubelets (the reference's scale
tests run pods on simulated nodes, benchmark/README.md): a bound pod
starts Running on tick; completion/failure is scripted per pod name.
"""

from __future__ import annotations

from typing import Callable, Optional

from ..store import ObjectStore


class FakeKubelet:
    def __init__(self, store: ObjectStore):
        self.store = store

    def tick(self, complete: Optional[Callable[[object], Optional[str]]] = None) -> int:
        """Advance pod phases: bound Pending→Running; `complete(pod)` may
        return 'Succeeded'/'Failed' to finish a Running pod."""
        n = 0
        for pod in self.store.list("Pod"):
            if pod.node_name and pod.phase == "Pending":
                pod.phase = "Running"
                self.store.update("Pod", pod)
                n += 1
            elif pod.phase == "Running" and complete is not None:
                outcome = complete(pod)
                if outcome in ("Succeeded", "Failed"):
                    pod.phase = outcome
                    self.store.update("Pod", pod)
                    n += 1
        return n
