"""Deviceshare plugin (reference ``plugins/deviceshare``): GPU sharing /
vGPU slicing through per-node device pools (api/devices.py).

The dense kernel handles the aggregate resource fit; this plugin adds
the per-card packing predicate: for each distinct device request shape,
nodes whose pool cannot pack the slices get a dynamic forbid bit; on
commit the pool assigns card indices (written into the task's pod
annotations — the ``volcano.sh/gpu-index`` contract the device plugin
on the node consumes)."""

from __future__ import annotations

from typing import Dict

from ...api.devices import ANN_ASSIGNED, DeviceRequest, GPUDevicePool
from ..tensors import set_plane_bit
from .base import Plugin, register


@register("deviceshare")
class DeviceSharePlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        nt = ssn.node_tensors
        # pools persist across cycles on the cache (usage carries over)
        pools: Dict[str, GPUDevicePool] = getattr(
            ssn.cache, "_device_pools", None) or {}
        for name, ni in ssn.nodes.items():
            if name not in pools:
                pool = GPUDevicePool(ni.node)
                if pool.cards:
                    pools[name] = pool
        ssn.cache._device_pools = pools
        if not pools:
            return
        self.pools = pools
        unfit_bits: Dict[tuple, int] = {}

        def request_of(tclass) -> DeviceRequest:
            pod = tclass.tasks[0].pod
            if pod is None:
                return None
            return DeviceRequest.from_annotations(pod.meta.annotations)

        def hook(tclass, job, require, forbid):
            req = request_of(tclass)
            if req is None:
                return
            sig = req.signature()
            bit = unfit_bits.get(sig)
            if bit is None:
                unfit = []
                for name, ni in ssn.nodes.items():
                    pool = pools.get(name)
                    if pool is None or pool.fit(req) is None:
                        unfit.append(ni.node_id)
                bit = nt.add_dynamic_bit(f"dev-unfit:{sig}", unfit)
                unfit_bits[sig] = bit
            set_plane_bit(forbid, bit)

        node_by_id = {ni.node_id: ni for ni in ssn.nodes.values()}

        def on_allocate(tclass, node_ids, counts, tasks=None):
            req = request_of(tclass)
            if req is None:
                return
            it = iter(tasks if tasks is not None else tclass.tasks)
            for nid, cnt in zip(node_ids, counts):
                ni = node_by_id.get(nid)
                pool = pools.get(ni.name) if ni else None
                for _ in range(cnt):
                    task = next(it, None)
                    if task is None:
                        return
                    picked = pool.allocate(req) if pool else None
                    if picked is not None and task.pod is not None:
                        task.pod.meta.annotations[ANN_ASSIGNED] = \
                            ",".join(str(i) for i in picked)

        handler = type("DevHandler", (), {"on_allocate":
                                          staticmethod(on_allocate)})()
        ssn.class_constraint_hooks.append(hook)
        ssn.event_handlers.append(handler)

        # GPU binpack/spread node scoring (reference deviceshare.go:233-285
        # NodeOrder): classes carrying a device request get a bias plane
        # favoring fuller (binpack, default) or emptier (spread) nodes'
        # device pools — per-card fill averaged per node.
        import numpy as np
        mode = str(self.args.get("deviceshare.schedule-policy", "binpack"))
        weight = float(self.args.get("deviceshare.gpu-score-weight", 0.05))
        if weight > 0:
            N = nt.n
            fill = np.zeros(N, dtype=np.float32)
            pooled = np.zeros(N, dtype=bool)
            for name, pool in pools.items():
                ni = ssn.nodes.get(name)
                if ni is None or ni.node_id < 0 or not pool.cards:
                    continue
                f = sum((c.mem_used / max(c.mem_total, 1)) if not c.exclusive
                        else 1.0 for c in pool.cards) / len(pool.cards)
                fill[ni.node_id] = f
                pooled[ni.node_id] = True
            bias_row = (weight * fill if mode == "binpack"
                        else -weight * fill).astype(np.float32)
            bias_row[~pooled] = 0.0

            def gpu_bias(tclass, job):
                return bias_row if request_of(tclass) is not None else None

            ssn.class_bias_fns.append(gpu_bias)
