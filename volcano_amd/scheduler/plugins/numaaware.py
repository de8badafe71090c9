"""Numaaware plugin (reference ``plugins/numaaware``: topology-manager
policies none/best-effort/restricted/single-numa-node against the
Numatopology CRD; cpuset assignment providers).

Pods opt in with the annotation ``volcano.sh/numa-topology-policy``
("single-numa-node"/"restricted"/"best-effort").  single-numa-node /
restricted: the pod's CPU request must fit inside one NUMA zone's free
capacity — nodes that can't are a dynamic forbid bit per request shape;
on commit the chosen zone is pinned via the ``volcano.sh/numa-node``
annotation and the zone's usage tracked across cycles."""

from __future__ import annotations

from typing import Dict, List

from ...api.resource import CPU
from ..tensors import set_plane_bit
from .base import Plugin, register

ANN_POLICY = "volcano.sh/numa-topology-policy"
ANN_PINNED = "volcano.sh/numa-node"
ANN_CPUSET = "volcano.sh/cpuset"


class CpusetProvider:
    """cpuset assignment provider (reference ``numaaware/provider``):
    hands out SPECIFIC CPU ids from a pinned zone's pool — the contract
    the node-side cpumanager enforces via cgroup cpuset.cpus.  Whole
    cores only (ceil of millicores/1000); released ids return to the
    pool (cache._release_devices analog is the pod-delete path)."""

    def __init__(self, topo):
        self.pools: Dict[int, List[int]] = {}
        for z in topo.zones:
            cpus = list(z.cpus) if z.cpus else list(range(
                z.id * 1000, z.id * 1000 + int(z.cpu_milli // 1000)))
            self.pools[z.id] = cpus

    def assign(self, zone: int, cpu_milli: float) -> List[int]:
        need = max(1, int(-(-cpu_milli // 1000)))     # ceil cores
        pool = self.pools.get(zone, [])
        if len(pool) < need:
            return []
        out, self.pools[zone] = pool[:need], pool[need:]
        return out

    def release(self, zone: int, cpus: List[int]) -> None:
        self.pools.setdefault(zone, []).extend(cpus)


class NumaState:
    """Per-node free millicores per NUMA zone (persists on the cache)."""

    def __init__(self, topo):
        self.free: List[float] = [z.cpu_milli for z in topo.zones]
        self.cpuset = CpusetProvider(topo)

    def best_zone(self, cpu_milli: float):
        best = None
        for i, f in enumerate(self.free):
            if f + 0.1 >= cpu_milli and (best is None
                                         or f < self.free[best]):
                best = i
        return best

    def take(self, zone: int, cpu_milli: float) -> None:
        self.free[zone] -= cpu_milli

    def give(self, zone: int, cpu_milli: float) -> None:
        self.free[zone] += cpu_milli


@register("numaaware")
class NumaAwarePlugin(Plugin):
    def on_session_open(self, ssn) -> None:
        from ...utils.features import enabled
        if not enabled("ResourceTopology"):
            return
        store = getattr(ssn.cache, "store", None)
        topos = store.list("Numatopology") if store is not None else []
        if not topos:
            return
        nt = ssn.node_tensors
        states: Dict[str, NumaState] = getattr(ssn.cache, "_numa_states", {})
        for topo in topos:
            if topo.meta.name not in states:
                states[topo.meta.name] = NumaState(topo)
        ssn.cache._numa_states = states

        unfit_bits: Dict[float, int] = {}

        def policy_of(tclass) -> str:
            pod = tclass.tasks[0].pod
            return pod.meta.annotations.get(ANN_POLICY, "") if pod else ""

        def hook(tclass, job, require, forbid):
            policy = policy_of(tclass)
            if policy not in ("single-numa-node", "restricted"):
                return
            cpu = tclass.request.get(CPU)
            bit = unfit_bits.get(cpu)
            if bit is None:
                unfit = []
                for name, ni in ssn.nodes.items():
                    st = states.get(name)
                    if st is None or st.best_zone(cpu) is None:
                        unfit.append(ni.node_id)
                bit = nt.add_dynamic_bit(f"numa-unfit:{cpu}", unfit)
                unfit_bits[cpu] = bit
            set_plane_bit(forbid, bit)

        node_by_id = {ni.node_id: ni for ni in ssn.nodes.values()}

        def on_allocate(tclass, node_ids, counts, tasks=None):
            if policy_of(tclass) not in ("single-numa-node", "restricted",
                                         "best-effort"):
                return
            cpu = tclass.request.get(CPU)
            it = iter(tasks if tasks is not None else tclass.tasks)
            for nid, cnt in zip(node_ids, counts):
                ni = node_by_id.get(nid)
                st = states.get(ni.name) if ni else None
                for _ in range(cnt):
                    task = next(it, None)
                    if task is None:
                        return
                    if st is None:
                        continue
                    zone = st.best_zone(cpu)
                    if zone is not None:
                        st.take(zone, cpu)
                        if task.pod is not None:
                            ann = task.pod.meta.annotations
                            ann[ANN_PINNED] = str(zone)
                            ids = st.cpuset.assign(zone, cpu)
                            if ids:
                                ann[ANN_CPUSET] = ",".join(
                                    str(c) for c in ids)

        handler = type("NumaHandler", (), {"on_allocate":
                                           staticmethod(on_allocate)})()
        ssn.class_constraint_hooks.append(hook)
        ssn.event_handlers.append(handler)
