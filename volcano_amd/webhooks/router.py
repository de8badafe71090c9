"""Admission router (reference ``pkg/webhooks/router/``: AdmissionService
registry at paths like /jobs/mutate, served by webhook-manager and called
by the apiserver admission chain).

Here the chain sits in front of the object store: ``AdmissionChain``
wraps create/update, running every registered mutator then validator for
the kind; the HTTP exposure (apiserver) serves the same handlers at the
same paths.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Callable, List, Optional

from ..store import ObjectStore


class AdmissionError(Exception):
    """Admission rejection (reference: AdmissionResponse.Allowed=false)."""


@dataclass
class AdmissionService:
    path: str                      # e.g. "/jobs/validate"
    kind: str                      # object kind it applies to
    func: Callable                 # func(store, obj, op) -> None (may mutate)
    operations: tuple = ("CREATE", "UPDATE")


class AdmissionChain:
    def __init__(self, store: Optional[ObjectStore] = None):
        self.store = store
        self.services: List[AdmissionService] = []

    def register(self, svc: AdmissionService) -> None:
        self.services.append(svc)

    def admit(self, kind: str, obj, op: str = "CREATE") -> object:
        """Run mutators (…/mutate) then validators (…/validate)."""
        for phase in ("/mutate", "/validate"):
            for svc in self.services:
                if svc.kind == kind and op in svc.operations \
                        and svc.path.endswith(phase):
                    svc.func(self.store, obj, op)
        return obj

    # -- store wiring ---------------------------------------------------------
    def guard(self, store: ObjectStore) -> "GuardedStore":
        self.store = store
        return GuardedStore(store, self)


class GuardedStore:
    """ObjectStore facade with the admission chain on the write path —
    the apiserver role: every create/update passes webhooks first."""

    def __init__(self, store: ObjectStore, chain: AdmissionChain):
        self._store = store
        self._chain = chain

    def create(self, kind: str, obj):
        self._chain.admit(kind, obj, "CREATE")
        return self._store.create(kind, obj)

    def update(self, kind: str, obj):
        self._chain.admit(kind, obj, "UPDATE")
        return self._store.update(kind, obj)

    def delete(self, kind: str, namespace: str, name: str):
        obj = self._store.get(kind, namespace, name)
        if obj is not None:
            self._chain.admit(kind, obj, "DELETE")
        return self._store.delete(kind, namespace, name)

    def apply(self, kind: str, obj):
        op = "UPDATE" if self._store.get(
            kind, obj.meta.namespace, obj.meta.name) else "CREATE"
        self._chain.admit(kind, obj, op)
        return self._store.apply(kind, obj)

    def __getattr__(self, name):
        return getattr(self._store, name)


def default_chain(store: Optional[ObjectStore] = None) -> AdmissionChain:
    from .admissions import register_all
    chain = AdmissionChain(store)
    register_all(chain)
    return chain
