"""Nodegroup plugin (reference ``plugins/nodegroup/nodegroup.go``):
queue ↔ nodegroup affinity via the ``volcano.sh/nodegroup-name`` node
label.  Each queue's allowed/forbidden node set is projected into ONE
dynamic bit plane at session open, so the hard affinity check rides the
predicate kernel for free (require/forbid words per class)."""

from __future__ import annotations

from ...api.objects import LBL_NODEGROUP
from ..tensors import set_plane_bit
from .base import Plugin, register


@register("nodegroup")
class NodeGroupPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        # node id → group name
        group_of = {}
        for ni in ssn.nodes.values():
            g = ni.node.meta.labels.get(LBL_NODEGROUP)
            if g:
                group_of[ni.node_id] = g

        queue_bits = {}
        for q in ssn.queues.values():
            aff = q.queue.spec.affinity
            if not aff:
                continue
            required = set((aff.get("affinity") or {}).get("required", []))
            anti = set((aff.get("antiAffinity") or {}).get("required", []))
            req_bit = forb_bit = None
            if required:
                ids = [nid for nid, g in group_of.items() if g in required]
                req_bit = nt.add_dynamic_bit(f"qng-req:{q.name}", ids)
            if anti:
                ids = [nid for nid, g in group_of.items() if g in anti]
                forb_bit = nt.add_dynamic_bit(f"qng-anti:{q.name}", ids)
            queue_bits[q.name] = (req_bit, forb_bit)
        if not queue_bits:
            return

        def hook(tclass, job, require, forbid):
            if job is None:
                return
            bits = queue_bits.get(job.queue)
            if not bits:
                return
            req_bit, forb_bit = bits
            if req_bit is not None and req_bit // 64 < len(require):
                set_plane_bit(require, req_bit)
            if forb_bit is not None and forb_bit // 64 < len(forbid):
                set_plane_bit(forbid, forb_bit)

        ssn.class_constraint_hooks.append(hook)
