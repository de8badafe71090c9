"""Minimal 5-field cron expression matcher (reference uses robfig/cron).

Supports: ``*``, lists ``1,2,3``, ranges ``1-5``, steps ``*/15`` and
``2-10/2``.  Fields: minute hour day-of-month month day-of-week.
"""

from __future__ import annotations

import time
from typing import Optional, Set

_BOUNDS = [(0, 59), (0, 23), (1, 31), (1, 12), (0, 6)]


def _parse_field(spec: str, lo: int, hi: int) -> Set[int]:
    out: Set[int] = set()
    for part in spec.split(","):
        step = 1
        if "/" in part:
            part, s = part.split("/", 1)
            step = int(s)
        if part in ("*", ""):
            lo2, hi2 = lo, hi
        elif "-" in part:
            a, b = part.split("-", 1)
            lo2, hi2 = int(a), int(b)
        else:
            lo2 = hi2 = int(part)
        for v in range(lo2, hi2 + 1, step):
            if lo <= v <= hi:
                out.add(v)
    return out


class CronSchedule:
    def __init__(self, expr: str):
        fields = expr.split()
        if len(fields) != 5:
            raise ValueError(f"bad cron expression {expr!r}")
        self.fields = [_parse_field(f, lo, hi)
                       for f, (lo, hi) in zip(fields, _BOUNDS)]

    def matches(self, t: Optional[float] = None) -> bool:
        tm = time.localtime(t if t is not None else time.time())
        minute, hour, dom, month, dow = (tm.tm_min, tm.tm_hour, tm.tm_mday,
                                         tm.tm_mon, tm.tm_wday)
        dow = (dow + 1) % 7          # python: Mon=0; cron: Sun=0
        return (minute in self.fields[0] and hour in self.fields[1]
                and dom in self.fields[2] and month in self.fields[3]
                and dow in self.fields[4])

    def next_after(self, t: float, horizon_s: int = 366 * 24 * 3600) -> Optional[float]:
        """Next matching minute strictly after t (minute resolution)."""
        start = int(t // 60 + 1) * 60
        for m in range(start, start + horizon_s, 60):
            if self.matches(m):
                return float(m)
        return None
