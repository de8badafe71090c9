"""Scheduler configuration (reference ``pkg/scheduler/conf/`` + the YAML
actions/tiers format of ``util.go:38-51 DefaultSchedulerConf``)."""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, List

import yaml


@dataclass
class PluginOption:
    name: str
    arguments: Dict[str, object] = field(default_factory=dict)


@dataclass
class Tier:
    plugins: List[PluginOption] = field(default_factory=list)


@dataclass
class SchedulerConfiguration:
    actions: List[str] = field(default_factory=lambda: ["enqueue", "allocate", "backfill"])
    tiers: List[Tier] = field(default_factory=list)
    configurations: Dict[str, Dict[str, object]] = field(default_factory=dict)
    # reference conf `metrics:` section (type/address/interval) — feeds
    # node real-usage into the cache (cache.setMetricsData analog)
    metrics: Dict[str, object] = field(default_factory=dict)
    schedule_period: float = 1.0
    use_hip: bool = False          # decision plane on GPU (HIP kernels)
    device: str = "cpu"
    feature_gates: Dict[str, bool] = field(default_factory=dict)

    @classmethod
    def from_yaml(cls, text: str) -> "SchedulerConfiguration":
        """Parse the reference's scheduler conf YAML shape:

        actions: "enqueue, allocate, backfill"
        tiers:
        - plugins:
          - name: priority
          - name: gang
        """
        data = yaml.safe_load(text) or {}
        conf = cls()
        if "actions" in data:
            conf.actions = [a.strip() for a in str(data["actions"]).split(",")
                            if a.strip()]
        tiers = []
        for tier in data.get("tiers", []):
            opts = []
            for p in tier.get("plugins", []):
                opts.append(PluginOption(name=p["name"],
                                         arguments=p.get("arguments", {}) or {}))
            tiers.append(Tier(plugins=opts))
        if tiers:
            conf.tiers = tiers
        for c in data.get("configurations", []) or []:
            conf.configurations[c.get("name", "")] = c.get("arguments", {}) or {}
        if isinstance(data.get("metrics"), dict):
            conf.metrics = dict(data["metrics"])
        fg = data.get("feature_gates") or data.get("featureGates") or {}
        if isinstance(fg, dict):
            conf.feature_gates = {str(k): bool(v) for k, v in fg.items()}
        return conf


def default_config() -> SchedulerConfiguration:
    """Reference DefaultSchedulerConf (pkg/scheduler/util.go:38-51):
    actions enqueue,allocate,backfill; tier1 priority/gang/conformance,
    tier2 overcommit/drf/predicates/proportion/nodeorder/binpack."""
    conf = SchedulerConfiguration()
    conf.tiers = [
        Tier(plugins=[PluginOption("priority"), PluginOption("gang"),
                      PluginOption("conformance")]),
        Tier(plugins=[PluginOption("overcommit"), PluginOption("drf"),
                      PluginOption("predicates"), PluginOption("proportion"),
                      PluginOption("nodeorder"), PluginOption("binpack")]),
    ]
    return conf
