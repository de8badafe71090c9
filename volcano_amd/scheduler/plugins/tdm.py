"""TDM (time-division multiplexing) plugin (reference ``plugins/tdm``):
revocable nodes (label ``volcano.sh/revocable-zone``) are usable by
*preemptable* jobs during their active time window; outside the window
the plugin evicts preemptable tasks from them (VictimTasks → shuffle).

Arguments: {"tdm.revocable-zone.<zone>": "HH:MM-HH:MM", ...}
"""

from __future__ import annotations

import time
from typing import Dict, Set

from ...api.objects import LBL_REVOCABLE_ZONE
from ..tensors import set_plane_bit
from .base import Plugin, register


def _window_active(spec: str, now: float) -> bool:
    try:
        start, end = spec.split("-")
        tm = time.localtime(now)
        cur = tm.tm_hour * 60 + tm.tm_min
        sh, sm = (int(x) for x in start.strip().split(":"))
        eh, em = (int(x) for x in end.strip().split(":"))
        s, e = sh * 60 + sm, eh * 60 + em
        return s <= cur <= e if s <= e else (cur >= s or cur <= e)
    except Exception:
        return False


@register("tdm")
class TdmPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        now = time.time()
        windows: Dict[str, str] = {
            k[len("tdm.revocable-zone."):]: str(v)
            for k, v in self.args.items()
            if k.startswith("tdm.revocable-zone.")}

        revocable_ids = []
        active_ids = []
        self._revocable_nodes: Set[str] = set()
        for ni in ssn.nodes.values():
            zone = ni.node.meta.labels.get(LBL_REVOCABLE_ZONE)
            if not zone:
                continue
            revocable_ids.append(ni.node_id)
            self._revocable_nodes.add(ni.name)
            if zone in windows and _window_active(windows[zone], now):
                active_ids.append(ni.node_id)
        if not revocable_ids:
            return
        rev_bit = nt.add_dynamic_bit("tdm-revocable", revocable_ids)
        inactive = [i for i in revocable_ids if i not in set(active_ids)]
        inact_bit = nt.add_dynamic_bit("tdm-inactive", inactive) \
            if inactive else None

        def hook(tclass, job, require, forbid):
            t = tclass.tasks[0]
            if not t.preemptable:
                # non-preemptable workloads never land on revocable nodes
                set_plane_bit(forbid, rev_bit)
            elif inact_bit is not None:
                # preemptable ones only during the active window
                set_plane_bit(forbid, inact_bit)

        def victim_tasks(tasks):
            """Window closed → evict preemptable tasks on revocable nodes."""
            out = []
            for t in tasks:
                if not t.preemptable or t.node_name not in self._revocable_nodes:
                    continue
                ni = ssn.nodes.get(t.node_name)
                if ni is None:
                    continue
                zone = ni.node.meta.labels.get(LBL_REVOCABLE_ZONE)
                if zone in windows and not _window_active(windows[zone], now):
                    out.append(t)
                elif zone not in windows:
                    out.append(t)
            return out

        def preemptable(preemptor, candidates):
            # revocable-node tasks are always fair game for preemption
            return [v for v in candidates
                    if v.node_name in self._revocable_nodes and v.preemptable]

        ssn.class_constraint_hooks.append(hook)
        ssn.victim_tasks_fns.append(victim_tasks)
        ssn.preemptable_fns.append(preemptable)
