"""Snapshot → dense tensor packing (the Session's device-side state).

The reference deep-clones the whole world into per-cycle Go maps
(`pkg/scheduler/cache/cache.go:1481` Snapshot).  The MI355X design instead
keeps the scheduling state as dense tensors sized for HBM3E:

* node resource planes ``alloc/used/extra`` laid out **[R, N]**
  (resource-major) so a wavefront's 64 lanes read 64 consecutive node
  entries of one resource dim — fully coalesced HBM3E lines;
* label/selector matching precompiled to **bit planes** ``[W, N]`` (64
  (key=value) pairs per int64 word) — a predicate is two AND-compares per
  word instead of string matching (reference predicate_helper.go:45 walks
  Go label maps per (task, node));
* taints/tolerations as a single int64 bitmask per node + a per-class
  tolerated mask.

Bit registries are sticky across cycles so packed planes stay valid under
incremental updates.
"""

from __future__ import annotations

from typing import Dict, List, Optional, Tuple

import numpy as np
import torch

from ..api.info import NodeInfo, TaskInfo
from ..api.objects import Node, Taint, Toleration
from ..api.resource import ResourceDims

# Effects that hard-block scheduling (PreferNoSchedule does not).
_BLOCKING_EFFECTS = ("NoSchedule", "NoExecute")


class BitRegistry:
    """Sticky mapping of string keys to bit positions across W int64 words."""

    def __init__(self, max_words: int = 4):
        self.index: Dict[str, int] = {}
        self.max_words = max_words

    def bit(self, key: str) -> int:
        i = self.index.get(key)
        if i is None:
            i = len(self.index)
            if i >= self.max_words * 64:
                raise OverflowError(
                    f"bit registry overflow (> {self.max_words * 64} distinct keys)")
            self.index[key] = i
        return i

    @property
    def words(self) -> int:
        return max(1, (len(self.index) + 63) // 64)


def _to_signed64(x: int) -> int:
    """Wrap a Python uint64 bit pattern into the int64 value range."""
    x &= (1 << 64) - 1
    return x - (1 << 64) if x >= (1 << 63) else x


def _set_bit(arr: np.ndarray, col: int, bit: int) -> None:
    arr[bit // 64, col] |= np.int64(_to_signed64(1 << (bit % 64)))


def set_plane_bit(words: np.ndarray, bit: int) -> None:
    """Set one bit in a [W] require/forbid word vector (int64-safe)."""
    words[bit // 64] |= np.int64(_to_signed64(1 << (bit % 64)))


class NodeTensors:
    """Packed per-node planes; rebuilt (v1) or patched per cycle."""

    def __init__(self, dims: ResourceDims, device: str = "cpu",
                 label_words: int = 16):
        self.dims = dims
        self.device = device
        self.labels = BitRegistry(label_words)
        self.taints = BitRegistry(1)
        self.names: List[str] = []
        self.index: Dict[str, int] = {}
        # torch views (created by pack)
        self.alloc_t: Optional[torch.Tensor] = None     # [R, N] f32
        self.used_t: Optional[torch.Tensor] = None      # [R, N] f32
        self.extra_t: Optional[torch.Tensor] = None     # [R, N] f32 (releasing - pipelined)
        self.ready: Optional[torch.Tensor] = None       # [N] u8
        self.taint_mask: Optional[torch.Tensor] = None  # [N] i64
        self.planes_t: Optional[torch.Tensor] = None    # [W, N] i64

    @property
    def n(self) -> int:
        return len(self.names)

    @property
    def r(self) -> int:
        return len(self.dims)

    # -- bit helpers (shared with class-constraint extraction) ---------------
    def label_bit(self, key: str, value: str) -> int:
        return self.labels.bit(f"{key}={value}")

    def taint_bit(self, taint: Taint) -> int:
        return self.taints.bit(f"{taint.key}={taint.value}:{taint.effect}")

    def tolerated_mask(self, tolerations: List[Toleration]) -> int:
        """int64 mask (signed value range) of taint bits the tolerations
        cover."""
        mask = 0
        for name, bit in self.taints.index.items():
            kv, _, effect = name.rpartition(":")
            key, _, value = kv.partition("=")
            t = Taint(key=key, value=value, effect=effect)
            for tol in tolerations:
                if tol.tolerates(t):
                    mask |= 1 << bit
                    break
        return _to_signed64(mask)

    def selector_bits(self, selector: Dict[str, str],
                      affinity: Optional[dict]) -> Tuple[np.ndarray, np.ndarray]:
        """(require, forbid) [W] int64 words from nodeSelector + simplified
        node affinity ({"in": {k: [v...]}, "notIn": {k: [v...]}}).

        Reference: predicates plugin nodeaffinity filter
        (plugins/predicates/predicates.go:34-47).  NOTE single-valued `in`
        lists and `notIn` compile exactly; multi-valued `in` falls back to
        requiring the first value only at pack level — callers with
        multi-value affinity get the host-side slow path (Session keeps a
        per-class candidate mask hook for that).
        """
        W = max(self.labels.words, 1)
        require = np.zeros(W, dtype=np.int64)
        forbid = np.zeros(W, dtype=np.int64)
        for k, v in (selector or {}).items():
            _set_bit(require.reshape(W, 1), 0, self.label_bit(k, v))
        if affinity:
            for k, vals in (affinity.get("in") or {}).items():
                if len(vals) == 1:
                    _set_bit(require.reshape(W, 1), 0, self.label_bit(k, vals[0]))
            for k, vals in (affinity.get("notIn") or {}).items():
                for v in vals:
                    _set_bit(forbid.reshape(W, 1), 0, self.label_bit(k, v))
        return require, forbid

    # -- packing -------------------------------------------------------------
    def pack_dynamic(self, nodes: List[NodeInfo], ledger=None) -> bool:
        """Re-pack only the per-cycle-mutable planes (used/extra) when the
        static planes (allocatable/labels/taints/ready) are still valid —
        the common steady-state path: pod churn changes usage, node churn
        is rare (SURVEY §7 'snapshot cost' hard-part).  Returns False if a
        full pack is required.

        With a :class:`NodeLedger` (the steady state) this is a pure
        vectorized astype/transpose of the ledger planes — no per-node
        work at all."""
        if self.alloc_t is None or len(nodes) != self.n:
            return False
        if self.alloc_t.shape[0] != self.r:
            return False    # dim registry grew since the static pack
        if [ni.name for ni in nodes] != self.names:
            return False
        N, R = self.n, self.r
        if ledger is not None and ledger.n == N and ledger.width >= R:
            from ..api.ledger import PIPELINED, RELEASING, USED
            p = ledger.planes
            used = p[USED, :, :R].T.astype(np.float32)
            extra = (p[RELEASING, :, :R] - p[PIPELINED, :, :R]) \
                .T.astype(np.float32)
            dev = self.device
            self.used_t.copy_(torch.from_numpy(np.ascontiguousarray(used)).to(dev))
            self.extra_t.copy_(torch.from_numpy(np.ascontiguousarray(extra)).to(dev))
            return True
        used = np.zeros((N, R), dtype=np.float32)
        extra = np.zeros((N, R), dtype=np.float32)
        didx = self.dims.index
        for i, ni in enumerate(nodes):
            uq = ni.used.q
            if uq:
                for k, v in uq.items():
                    j = didx.get(k)
                    if j is not None:
                        used[i, j] = v
            for k, v in ni.releasing.q.items():
                j = didx.get(k)
                if j is not None:
                    extra[i, j] += v
            for k, v in ni.pipelined.q.items():
                j = didx.get(k)
                if j is not None:
                    extra[i, j] -= v
        dev = self.device
        self.used_t.copy_(torch.from_numpy(used.T.copy()).to(dev))
        self.extra_t.copy_(torch.from_numpy(extra.T.copy()).to(dev))
        return True

    def pack(self, nodes: List[NodeInfo], ledger=None) -> None:
        """Full (re)pack from host NodeInfos.

        Vectorized via numpy staging buffers then one H2D per plane; at
        N=50k this is ~a few MB — the per-cycle upload is microseconds of
        PCIe time."""
        names = [ni.name for ni in nodes]
        self.names = names
        self.index = {n: i for i, n in enumerate(names)}
        N = len(nodes)

        # first walk fixes the dim count and assigns any unseen label/taint
        # bits so R and W are final before arrays are sized
        for ni in nodes:
            for k in ni.allocatable.q:
                self.dims.add(k)
            for t in ni.node.taints:
                if t.effect in _BLOCKING_EFFECTS:
                    self.taint_bit(t)
            for k, v in ni.node.meta.labels.items():
                self.label_bit(k, v)
        R = self.r
        W = self.labels.words

        alloc = np.zeros((N, R), dtype=np.float32)
        used = np.zeros((N, R), dtype=np.float32)
        extra = np.zeros((N, R), dtype=np.float32)
        ready = np.zeros(N, dtype=np.uint8)
        taints = np.zeros(N, dtype=np.int64)
        planes = np.zeros((W, N), dtype=np.int64)

        didx = self.dims.index
        # synthetic unit dims: anti-affinity groups ("paa:") and host
        # ports ("hp:<port>") — every node offers exactly 1, so the
        # capacity kernel enforces at-most-one-per-node (k8s nodeports
        # filter) with zero predicate work.  "nvl:attach" models the
        # node's attachable-volume limit (k8s nodevolumelimits): default
        # 256, overridden per node by the volcano.sh/max-volumes
        # annotation.
        nvl_j = didx.get("nvl:attach")
        for name, j in didx.items():
            if name.startswith("paa:") or name.startswith("hp:"):
                alloc[:, j] = 1.0
        if nvl_j is not None:
            alloc[:, nvl_j] = 256.0
        use_ledger = ledger is not None and ledger.n == N \
            and ledger.width >= R
        if use_ledger:
            from ..api.ledger import PIPELINED, RELEASING, USED
            p = ledger.planes
            used[:] = p[USED, :, :R].astype(np.float32)
            extra[:] = (p[RELEASING, :, :R] - p[PIPELINED, :, :R]) \
                .astype(np.float32)
        for i, ni in enumerate(nodes):
            if nvl_j is not None:
                lim = ni.node.meta.annotations.get("volcano.sh/max-volumes")
                if lim:
                    alloc[i, nvl_j] = float(lim)
            for k, v in ni.allocatable.q.items():
                alloc[i, didx[k]] = v
            if not use_ledger:
                for k, v in ni.used.q.items():
                    used[i, didx[k]] = v
                for k, v in ni.releasing.q.items():
                    extra[i, didx[k]] += v
                for k, v in ni.pipelined.q.items():
                    extra[i, didx[k]] -= v
            ready[i] = 1 if ni.ready else 0
            for t in ni.node.taints:
                if t.effect in _BLOCKING_EFFECTS:
                    taints[i] |= np.int64(_to_signed64(1 << self.taint_bit(t)))
            for k, v in ni.node.meta.labels.items():
                _set_bit(planes, i, self.label_bit(k, v))

        dev = self.device
        # numpy mirrors for host-side rare paths (preempt/reclaim scans) —
        # avoids per-element D2H reads when the planes live on the GPU
        self.ready_np = ready
        self.taints_np = taints
        self.planes_np = planes
        # resource planes transposed to [R, N]: one wavefront reads 64
        # consecutive nodes of one dim (coalesced; see module docstring)
        self.alloc_t = torch.from_numpy(alloc.T.copy()).to(dev)
        self.used_t = torch.from_numpy(used.T.copy()).to(dev)
        self.extra_t = torch.from_numpy(extra.T.copy()).to(dev)
        self.ready = torch.from_numpy(ready).to(dev)
        self.taint_mask = torch.from_numpy(taints).to(dev)
        self.planes_t = torch.from_numpy(planes).to(dev)

    def req_vector(self, task: TaskInfo) -> Optional[np.ndarray]:
        """Dense request vector, or None if the task asks for a resource no
        node in the inventory offers (unschedulable this cycle)."""
        out = np.zeros(self.r, dtype=np.float32)
        for k, v in task.request.q.items():
            i = self.dims.index.get(k)
            if i is None:
                if v >= 0.1:
                    return None
                continue
            out[i] = v
        return out

    def clear_dynamic_bits(self) -> None:
        """Zero every dynamic bit across all nodes.  Called at session
        open: dynamic sets (device-unfit, usage-over, hypernode domains,
        …) are session state — plugins re-project them each cycle, and a
        stale membership from the previous cycle must not linger (a full
        pack clears them implicitly; incremental cycles do not)."""
        if self.planes_t is None or not getattr(self, "_dynamic_bits", None):
            return
        masks: Dict[int, int] = {}
        for bit in self._dynamic_bits:
            masks[bit // 64] = masks.get(bit // 64, 0) | (1 << (bit % 64))
        for w, m in masks.items():
            if w >= self.planes_np.shape[0]:
                continue
            inv = np.int64(_to_signed64(~m))
            self.planes_np[w] &= inv
            self.planes_t[w] &= int(inv)

    def add_dynamic_bit(self, name: str, node_ids) -> int:
        """Register (or reuse) a synthetic label bit and set it for the
        given node ids — plugins project arbitrary node sets (nodegroup
        membership, revocable zones, over-utilized nodes) into the SAME
        bit planes the predicate kernel already tests, so a new node-set
        predicate costs zero extra kernel work.  Returns the bit index."""
        bit = self.labels.bit(f"dyn:{name}")
        if not hasattr(self, "_dynamic_bits"):
            self._dynamic_bits = set()
        self._dynamic_bits.add(bit)
        w, b = bit // 64, bit % 64
        val = np.int64(_to_signed64(1 << b))
        if self.planes_t is None:
            return bit
        if w >= self.planes_np.shape[0]:
            pad_np = np.zeros((w + 1 - self.planes_np.shape[0], self.n),
                              dtype=np.int64)
            self.planes_np = np.concatenate([self.planes_np, pad_np])
            pad_t = torch.zeros(
                (w + 1 - self.planes_t.shape[0], self.planes_t.shape[1]),
                dtype=torch.int64, device=self.planes_t.device)
            self.planes_t = torch.cat([self.planes_t, pad_t])
        ids = np.asarray(list(node_ids), dtype=np.int64)
        if len(ids):
            self.planes_np[w, ids] |= val
            idx = torch.from_numpy(ids).to(self.planes_t.device)
            row = self.planes_t[w]
            row[idx] |= torch.tensor(int(val), dtype=torch.int64,
                                     device=row.device)
        return bit

    def ensure_plane_width(self) -> None:
        """Pad the packed planes to the registry's word count (a class
        selector can register bits for labels no node carries — those
        planes are all-zero and the class is simply infeasible there)."""
        if self.planes_t is None:
            return
        W = max(self.labels.words, 1)
        have = self.planes_t.shape[0]
        if have >= W:
            return
        pad_np = np.zeros((W - have, self.n), dtype=np.int64)
        self.planes_np = np.concatenate([self.planes_np, pad_np])
        pad_t = torch.zeros((W - have, self.planes_t.shape[1]),
                            dtype=torch.int64, device=self.planes_t.device)
        self.planes_t = torch.cat([self.planes_t, pad_t])

    def bit_words(self, bits) -> np.ndarray:
        """Pack bit indices into a [W] require/forbid word vector."""
        W = max(self.labels.words, 1)
        out = np.zeros(W, dtype=np.int64)
        for bit in bits:
            out[bit // 64] |= np.int64(_to_signed64(1 << (bit % 64)))
        return out

    def resource_vector(self, resource) -> np.ndarray:
        """Dense vector of a Resource over the current dims (unknown dims
        dropped — used for queue/cluster aggregates)."""
        out = np.zeros(self.r, dtype=np.float32)
        for k, v in resource.q.items():
            i = self.dims.index.get(k)
            if i is not None:
                out[i] = v
        return out
