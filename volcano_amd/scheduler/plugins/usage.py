"""Usage plugin (reference ``plugins/usage``): real-utilization-aware
scheduling — nodes whose measured usage exceeds thresholds are filtered
(dynamic forbid bit through the predicate kernel); a usage-based score
bias steers load to idle nodes (the global bias plane of the fused
score kernel — K2).

Usage source: node annotations ``volcano.sh/cpu-usage`` /
``volcano.sh/memory-usage`` (0-100), the in-process analog of the
reference's Prometheus/Elasticsearch metrics clients
(``pkg/scheduler/metrics/source``) — the node agent publishes them.
"""

from __future__ import annotations

import numpy as np
import torch

from ..tensors import set_plane_bit
from .base import Plugin, register

ANN_CPU_USAGE = "volcano.sh/cpu-usage"
ANN_MEM_USAGE = "volcano.sh/memory-usage"


@register("usage")
class UsagePlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        cpu_thresh = float(self.args.get("thresholds", {}).get("cpu", 80)) \
            if isinstance(self.args.get("thresholds"), dict) \
            else float(self.args.get("usage.cpu-threshold", 80))
        mem_thresh = float(self.args.get("thresholds", {}).get("mem", 80)) \
            if isinstance(self.args.get("thresholds"), dict) \
            else float(self.args.get("usage.memory-threshold", 80))
        weight = float(self.args.get("usage.weight", 5))

        # pluggable usage source (reference metrics/source MetricsClient):
        # annotations by default, prometheus via arguments
        from ..metrics_source import new_source
        kind = self.args.get("usage.metrics-source", "annotation")
        src_kw = {}
        if kind == "prometheus":
            src_kw["address"] = self.args.get("usage.prometheus-address",
                                              "http://127.0.0.1:9090")
        source = new_source(kind, **src_kw)
        usage = source.node_usage([ni.node for ni in ssn.nodes.values()])

        N = nt.n
        cpu = np.zeros(N, dtype=np.float32)
        mem = np.zeros(N, dtype=np.float32)
        for ni in ssn.nodes.values():
            u = usage.get(ni.name, {})
            cpu[ni.node_id] = u.get("cpu", 0.0)
            mem[ni.node_id] = u.get("memory", 0.0)

        over = np.nonzero((cpu > cpu_thresh) | (mem > mem_thresh))[0]
        if len(over):
            bit = nt.add_dynamic_bit("usage-over", over.tolist())

            def hook(tclass, job, require, forbid):
                set_plane_bit(forbid, bit)

            ssn.class_constraint_hooks.append(hook)

        # score bias: prefer low measured usage (normalized to ~[0, weight])
        bias = weight * (1.0 - (cpu + mem) / 200.0)
        prev = getattr(ssn, "score_bias", None)
        b = torch.from_numpy(bias)
        ssn.score_bias = b if prev is None else prev + b
