"""Columnar node accounting — the host-side usage planes.

Round-1 profiling showed the scheduler cycle host-bound: per-node
``Resource`` dict math (12.5k ``add_allocated_bulk`` calls per 100k-pod
cycle) and the per-node dict walks in the tensor re-pack dominated the
step.  The ledger replaces the per-``NodeInfo`` Resource objects as the
truth for used/releasing/pipelined with dense ``[N, R]`` float64 arrays
aligned to the packed node order, so:

* the allocate commit is ONE ``np.add.at`` per task class instead of one
  Resource op per (node, class) piece;
* the per-cycle tensor re-pack is a vectorized ``astype/transpose`` of
  the ledger arrays instead of a 10k-node dict walk;
* bench/churn resets are ``arr[:] = 0``.

``NodeInfo`` keeps its ``used``/``releasing``/``pipelined`` API — the
properties materialize a ``Resource`` snapshot from the row on access
(cold paths: preempt victim scans, dump, tests).  Mutations MUST go
through ``NodeInfo`` methods (add_task/remove_task/...) or ledger bulk
ops; materialized snapshots are read-only views by convention.

Reference analog: the deep-cloned ``NodeInfo.Idle/Used/Releasing``
Resources rebuilt every cycle by ``cache.Snapshot``
(/root/reference/pkg/scheduler/cache/cache.go:1481,
pkg/scheduler/api/node_info.go:52-97).
"""

from __future__ import annotations

from typing import Dict, List

import numpy as np

from .resource import Resource, ResourceDims

USED, RELEASING, PIPELINED, REMOTE = 0, 1, 2, 3


class NodeLedger:
    """Dense [N, R] float64 usage planes over the packed node order."""

    def __init__(self, dims: ResourceDims, n: int):
        self.dims = dims
        cap = max(len(dims), 8)
        self.n = n
        # planes[k] : [N, cap] — k in (USED, RELEASING, PIPELINED, REMOTE)
        self.planes = np.zeros((4, n, cap), dtype=np.float64)
        self.alloc = np.zeros((n, cap), dtype=np.float64)
        self.version = 0

    @property
    def width(self) -> int:
        return self.planes.shape[2]

    def _widen(self, need: int) -> None:
        cap = max(need, self.width * 2)
        pad = cap - self.width
        self.planes = np.pad(self.planes, ((0, 0), (0, 0), (0, pad)))
        self.alloc = np.pad(self.alloc, ((0, 0), (0, pad)))

    # -- scalar (cold-path) ops ---------------------------------------------
    def apply(self, row: int, q: Dict[str, float],
              du: float = 0.0, dr: float = 0.0, dp: float = 0.0) -> None:
        """Accumulate a Resource dict into one node row (sign-scaled)."""
        didx = self.dims.index
        p = self.planes
        for k, v in q.items():
            i = didx.get(k)
            if i is None:
                i = self.dims.add(k)
            if i >= self.width:
                self._widen(i + 1)
                p = self.planes
            if du:
                p[USED, row, i] += du * v
            if dr:
                p[RELEASING, row, i] += dr * v
            if dp:
                p[PIPELINED, row, i] += dp * v
        self.version += 1

    def set_row(self, row: int, plane: int, r: Resource) -> None:
        self.planes[plane, row, :] = 0.0
        didx = self.dims.index
        for k, v in r.q.items():
            i = didx.get(k)
            if i is None:
                i = self.dims.add(k)
            if i >= self.width:
                self._widen(i + 1)
            self.planes[plane, row, i] = v
        self.version += 1

    def resource(self, row: int, plane: int) -> Resource:
        """Materialize a read-only Resource snapshot of one row."""
        r = self.planes[plane, row]
        names = self.dims.names
        nz = np.nonzero(r)[0]
        return Resource({names[i]: float(r[i]) for i in nz if i < len(names)})

    def row_vec(self, row: int, plane: int) -> np.ndarray:
        return self.planes[plane, row]

    # -- bulk (hot-path) ops --------------------------------------------------
    def add_used_bulk(self, rows: np.ndarray, counts: np.ndarray,
                      req: np.ndarray) -> None:
        """used[rows] += counts[:,None] * req — one call per task class.
        Duplicate rows accumulate (np.add.at semantics)."""
        w = self.width
        if len(req) > w:
            self._widen(len(req))
            w = self.width
        vals = counts[:, None].astype(np.float64) * \
            req[None, :].astype(np.float64)
        np.add.at(self.planes[USED, :, :len(req)], rows, vals)
        self.version += 1

    def add_used_rows(self, rows: np.ndarray, vals: np.ndarray) -> None:
        """used[rows] += vals ([K, R] per-piece deltas — the whole apply
        pass in ONE np.add.at; duplicate rows accumulate)."""
        w = self.width
        R = vals.shape[1]
        if R > w:
            self._widen(R)
        np.add.at(self.planes[USED, :, :R], rows, vals)
        self.version += 1

    def zero_usage(self) -> None:
        self.planes[:] = 0.0
        self.version += 1

    # -- (re)build -------------------------------------------------------------
    @classmethod
    def build(cls, dims: ResourceDims, nodes: List) -> "NodeLedger":
        """Adopt `nodes` (name-sorted packed order).  Values come from each
        node's current truth — its old ledger row if adopted, else its
        local pre-adoption Resources."""
        led = cls(dims, len(nodes))
        didx = dims.index
        for i, ni in enumerate(nodes):
            old = ni._ledger
            if old is not None and ni._row >= 0:
                w = min(old.width, led.width)
                led.planes[:, i, :w] = old.planes[:, ni._row, :w]
            else:
                for plane, r in ((USED, ni._used), (RELEASING, ni._releasing),
                                 (PIPELINED, ni._pipelined),
                                 (REMOTE, ni._remote_used)):
                    for k, v in r.q.items():
                        j = didx.get(k)
                        if j is None:
                            j = dims.add(k)
                        if j >= led.width:
                            led._widen(j + 1)
                        led.planes[plane, i, j] = v
            for k, v in ni.allocatable.q.items():
                j = didx.get(k)
                if j is None:
                    j = dims.add(k)
                if j >= led.width:
                    led._widen(j + 1)
                led.alloc[i, j] = v
            ni._ledger = led
            ni._row = i
        return led
