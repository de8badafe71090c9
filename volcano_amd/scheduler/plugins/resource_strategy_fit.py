"""Resource-strategy-fit plugin (reference
``plugins/resource-strategy-fit``): per-resource binpack-vs-spread
("MostAllocated"/"LeastAllocated") with per-resource weights.

MI355X mapping: the fused score kernel computes
``score = w_most·Σ_r dim_w_r·f_r / Σdim_w + …`` — a per-resource
*direction* is expressed with SIGNED dim weights (+w binpack, −w spread;
argmax is invariant to the constant and positive scale this introduces),
so no extra kernel is needed.

Arguments: {"resources": {"cpu": {"type": "MostAllocated", "weight": 2},
                          "memory": {"type": "LeastAllocated", "weight": 1}}}
"""

from __future__ import annotations

from .base import Plugin, register


@register("resource-strategy-fit")
class ResourceStrategyFitPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        resources = self.args.get("resources", {}) or {}
        if not resources:
            return
        # this plugin owns the requested-fraction terms: the symmetric
        # least term would cancel the signed weights (least+most ≡ const)
        ssn.score_weights["most"] = 1.0
        ssn.score_weights["least"] = 0.0
        for name, spec in resources.items():
            w = float(spec.get("weight", 1.0))
            if spec.get("type") == "LeastAllocated":
                w = -w
            ssn.dim_weights[name] = w
        # dims the config doesn't mention don't contribute
        nt = ssn.node_tensors
        for name in nt.dims.names:
            ssn.dim_weights.setdefault(name, 0.0)
