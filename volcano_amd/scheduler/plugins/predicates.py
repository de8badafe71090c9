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

    def class_constraints(self, tclass,
                          job=None) -> Tuple[int, np.ndarray, np.ndarray]:
        """(tolerated taint mask, require planes, forbid planes) for a class.

        All tasks of a class share constraints by construction
        (TaskInfo.class_signature); memoized per (signature, queue) —
        thousands of identical gangs share one entry.  Other plugins
        (nodegroup/tdm/usage) extend the bits through
        ``ssn.class_constraint_hooks``."""
        hooks = getattr(self._ssn, "class_constraint_hooks", [])
        key = (tclass.signature, job.queue if (job and hooks) else None)
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
        extra_bits = self._multi_value_in_bits(pod)
        require, forbid = nt.selector_bits(
            pod.node_selector if pod else {}, pod.affinity if pod else None)
        if extra_bits:
            from ..tensors import set_plane_bit
            for bit in extra_bits:
                set_plane_bit(require, bit)
        for hook in hooks:
            hook(tclass, job, require, forbid)
        out = (tolerated, require, forbid)
        self._memo[key] = out
        return out

    def _multi_value_in_bits(self, pod):
        """Multi-value nodeAffinity ``In`` lists are an OR over label
        values — not expressible as an AND-mask directly, so each matching
        node SET becomes one dynamic require bit (memoized)."""
        aff = pod.affinity if pod is not None else None
        if not isinstance(aff, dict):
            return ()
        nt = self._nt
        bits = []
        for k, vals in (aff.get("in") or {}).items():
            if len(vals) <= 1:
                continue       # single-value handled by selector_bits
            memo_key = ("in", k, tuple(sorted(vals)))
            bit = self._memo.get(memo_key)
            if bit is None:
                vset = set(vals)
                ids = [ni.node_id for ni in self._ssn.nodes.values()
                       if ni.node.meta.labels.get(k) in vset]
                bit = nt.add_dynamic_bit(
                    f"selin:{k}:{','.join(sorted(vals))}", ids)
                self._memo[memo_key] = bit
            bits.append(bit)
        return bits
