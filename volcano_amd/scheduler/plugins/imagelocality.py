"""Image-locality scoring (k8s imagelocality score plugin, wrapped by
reference ``plugins/nodeorder`` — nodeorder.go:34-66 weight table).

Nodes that already hold a pod's container image avoid the pull: they get
an additive per-class score bias (the same kernel bias-plane input
task-topology bucket packing uses), so image preference is evaluated for
ALL nodes inside the fused score pass — no per-(task,node) callbacks.
The bias plane for one image is built once per session and SHARED across
every class requesting that image (deduplicated device upload)."""

from __future__ import annotations

from typing import Dict

import numpy as np

from .base import Plugin, register


@register("imagelocality")
class ImageLocalityPlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        weight = float(self.args.get("imagelocality.weight", 5.0))
        nt = ssn.node_tensors
        planes: Dict[str, object] = {}   # image -> shared bias row (or None)

        def plane_for(image):
            if image in planes:
                return planes[image]
            bias = None
            for ni in ssn.nodes.values():
                if ni.node_id >= 0 and image in ni.node.images:
                    if bias is None:
                        bias = np.zeros(nt.n, dtype=np.float32)
                    bias[ni.node_id] = weight
            planes[image] = bias
            return bias

        def class_bias(tclass, job):
            if nt is None or nt.n == 0:
                return None
            pod = tclass.tasks[0].pod
            if pod is None or not pod.image:
                return None
            return plane_for(pod.image)

        ssn.class_bias_fns.append(class_bias)
