"""Predicates plugin (reference ``plugins/predicates/predicates.go``).

The reference wraps the k8s in-tree filter plugins (nodeunschedulable,
nodeaffinity, tainttoleration, ...) as per-(task,node) Go callbacks run
16-way parallel over a 5 % node sample (util/predicate_helper.go:45,
scheduler_helper.go:56).  The MI355X design precompiles the same
constraints to per-node bit planes and masks at session open; the fused
score kernel (K1, ops/csrc/scheduler_kernels.hip) evaluates them for ALL
nodes per class in one pass — no sampling, strictly better placements.

This plugin owns turning each task class's pod constraints into
(tolerated_mask, require_bits, forbid_bits); the allocate action calls
``class_constraints``.
"""

from __future__ import annotations

from typing import Tuple

import numpy as np

from .base import Plugin, register


@register("predicates")
class PredicatesPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        ssn.predicates = self
        self._ssn = ssn
        self._nt = ssn.node_tensors
        self._memo = {}

    def class_constraints(self, tclass, job=None,
                          skip_topology=False
                          ) -> Tuple[int, np.ndarray, np.ndarray]:
        """(tolerated taint mask, require planes, forbid planes) for a class.

        All tasks of a class share constraints by construction
        (TaskInfo.class_signature); memoized per (signature, queue) —
        thousands of identical gangs share one entry.  Other plugins
        (nodegroup/tdm/usage) extend the bits through
        ``ssn.class_constraint_hooks``."""
        hooks = getattr(self._ssn, "class_constraint_hooks", [])
        if skip_topology:
            # hypernode-domain preemption trials restrict nodes via an
            # explicit domain filter; the topology hook's own domain
            # choice (made BEFORE the trial's evictions) must not leak in
            hooks = [h for h in hooks
                     if not getattr(h, "is_topology", False)]
        # per-class topology overrides (SubGroupPolicy) choose a domain
        # PER SUBGROUP — never share a memo entry across subgroups
        memoize = getattr(tclass, "topology", None) is None \
            and not skip_topology
        key = (tclass.signature, job.queue if (job and hooks) else None)
        if memoize:
            got = self._memo.get(key)
            if got is not None:
                return got
        nt = self._nt
        t = tclass.tasks[0]
        pod = t.pod
        tolerations = pod.tolerations if pod else []
        tolerated = nt.tolerated_mask(tolerations)
        # register OR-set bits BEFORE the word arrays are sized so the
        # widths line up
        extra_req, extra_forbid = self._affinity_bits(pod)
        extra_req = tuple(extra_req) + self._volume_zone_bits(pod)
        require, forbid = nt.selector_bits(
            pod.node_selector if pod else {}, pod.affinity if pod else None)
        if extra_req or extra_forbid:
            from ..tensors import set_plane_bit
            for bit in extra_req:
                set_plane_bit(require, bit)
            for bit in extra_forbid:
                set_plane_bit(forbid, bit)
        for hook in hooks:
            hook(tclass, job, require, forbid)
        out = (tolerated, require, forbid)
        if memoize:
            self._memo[key] = out
        return out

    def _volume_zone_bits(self, pod) -> tuple:
        """Volume-zone filter (k8s volumezone, wrapped by reference
        plugins/predicates): a pod consuming a bound PVC whose PV carries
        a zone label must land on nodes of that zone.  Unbound claims
        constrain nothing (WaitForFirstConsumer).  Zones ride the static
        label bit planes — zero kernel cost."""
        if pod is None or not pod.volumes:
            return ()
        from ...utils.features import enabled
        if not enabled("CSIStorage"):
            return ()
        store = getattr(self._ssn.cache, "store", None)
        if store is None:
            return ()
        from ...api.objects import ZONE_LABEL
        key = ("volzone", pod.meta.namespace, tuple(sorted(pod.volumes)))
        got = self._memo.get(key)
        if got is not None:
            return got
        pv_index = self._memo.get("_pv_index")
        if pv_index is None:
            pv_index = self._memo["_pv_index"] = {
                pv.meta.name: pv
                for pv in store.list("PersistentVolume")}
        bits = []
        for vname in pod.volumes:
            pvc = store.get("PersistentVolumeClaim", pod.meta.namespace,
                            vname)
            if pvc is None or not pvc.volume_name:
                continue
            pv = pv_index.get(pvc.volume_name)
            if pv is not None and pv.zone:
                bits.append(self._nt.label_bit(ZONE_LABEL, pv.zone))
        got = self._memo[key] = tuple(bits)
        return got

    def _node_set_bit(self, memo_key, name, match):
        """Memoized dynamic require/forbid bit for the node set where
        ``match(labels)`` holds."""
        bit = self._memo.get(memo_key)
        if bit is None:
            ids = [ni.node_id for ni in self._ssn.nodes.values()
                   if match(ni.node.meta.labels)]
            bit = self._nt.add_dynamic_bit(name, ids)
            self._memo[memo_key] = bit
        return bit

    def _affinity_bits(self, pod):
        """Node-affinity operators beyond exact-match AND: each compiles
        to a dynamic bit over its matching node SET (memoized per
        expression).  Covers the k8s NodeSelectorOperator surface
        (k8s.io/api/core/v1 In/NotIn/Exists/DoesNotExist/Gt/Lt wrapped by
        reference plugins/predicates/predicates.go nodeaffinity):

        * ``in`` with >1 value — OR over values → one require bit;
        * ``exists: [key, ...]`` — label present → require bit per key;
        * ``notExists: [key, ...]`` — label present → forbid bit;
        * ``gt/lt: {key: num}`` — numeric label compare → require bit.
        """
        aff = pod.affinity if pod is not None else None
        if not isinstance(aff, dict):
            return (), ()
        req, forbid = [], []

        def num(x):
            try:
                return float(x)
            except (TypeError, ValueError):
                return None

        for k, vals in (aff.get("in") or {}).items():
            if len(vals) <= 1:
                continue       # single-value handled by selector_bits
            vset = set(vals)
            req.append(self._node_set_bit(
                ("in", k, tuple(sorted(vals))),
                f"selin:{k}:{','.join(sorted(vals))}",
                lambda lbl, k=k, vset=vset: lbl.get(k) in vset))
        for k in (aff.get("exists") or []):
            req.append(self._node_set_bit(
                ("exists", k), f"selex:{k}",
                lambda lbl, k=k: k in lbl))
        for k in (aff.get("notExists") or []):
            forbid.append(self._node_set_bit(
                ("exists", k), f"selex:{k}",
                lambda lbl, k=k: k in lbl))
        for k, v in (aff.get("gt") or {}).items():
            vf = float(v)
            req.append(self._node_set_bit(
                ("gt", k, vf), f"selgt:{k}:{v}",
                lambda lbl, k=k, vf=vf: (lambda n: n is not None and n > vf)(
                    num(lbl.get(k)))))
        for k, v in (aff.get("lt") or {}).items():
            vf = float(v)
            req.append(self._node_set_bit(
                ("lt", k, vf), f"sellt:{k}:{v}",
                lambda lbl, k=k, vf=vf: (lambda n: n is not None and n < vf)(
                    num(lbl.get(k)))))
        return req, forbid
