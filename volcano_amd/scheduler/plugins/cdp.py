"""CDP plugin (reference ``plugins/cdp`` — cooldown protection): freshly
started pods are protected from preemption for ``cooldown-time`` seconds
(arguments: {"cdp.cooldown-time": "300s"})."""

from __future__ import annotations

import time

from .base import Plugin, register
from .sla import _parse_duration


@register("cdp")
class CdpPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        cooldown = _parse_duration(self.args.get("cdp.cooldown-time", "0s"))
        if cooldown <= 0:
            return
        now = time.time()

        def protect(evictor, candidates):
            out = []
            for v in candidates:
                started = v.pod.meta.creation_timestamp if v.pod else 0.0
                if now - started >= cooldown:
                    out.append(v)
            return out

        ssn.preemptable_fns.append(protect)
