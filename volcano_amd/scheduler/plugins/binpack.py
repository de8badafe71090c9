"""Binpack plugin (reference ``plugins/binpack/binpack.go:193-247``).

``score_r = (used + request) / allocatable * weight_r`` summed over dims —
in the reference a per-(task,node) Go loop; here it IS the ``w_most`` term
of the fused score kernel (K2): the plugin just turns the weight on and
contributes per-dim weights.
"""

from __future__ import annotations

from .base import Plugin, register


@register("binpack")
class BinpackPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        weight = float(self.args.get("binpack.weight", 1.0))
        ssn.score_weights["most"] = ssn.score_weights.get("most", 0.0) + weight
        for key, val in self.args.items():
            if key.startswith("binpack.resources."):
                ssn.dim_weights[key[len("binpack.resources."):]] = float(val)
