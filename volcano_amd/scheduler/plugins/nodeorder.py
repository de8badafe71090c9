"""Nodeorder plugin (reference ``plugins/nodeorder/nodeorder.go:34-66``).

Weighted sum of leastrequested / mostrequested / balancedallocation — the
three analytic terms of the fused score kernel (K2).  Taint-toleration and
affinity hard filters are handled by the predicates plugin's bit planes;
the soft (preferred) variants contribute to the same score via the bias
plane when configured.
"""

from __future__ import annotations

from .base import Plugin, register


@register("nodeorder")
class NodeOrderPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        ssn.score_weights["least"] = float(
            self.args.get("leastrequested.weight", 1.0))
        ssn.score_weights["bal"] = float(
            self.args.get("balancedresource.weight", 1.0))
        most = float(self.args.get("mostrequested.weight", 0.0))
        if most:
            ssn.score_weights["most"] = ssn.score_weights.get("most", 0.0) + most
