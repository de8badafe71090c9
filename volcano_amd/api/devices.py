"""Per-node shareable device pools (reference ``pkg/scheduler/api/
devices/`` — the Devices interface shared_device_pool.go:34-70 with
nvidia/gpushare + nvidia/vgpu implementations; here the AMD analog).

Pod-side request annotations (reference keys, AMD-flavored):
  volcano.sh/gpu-number  — whole shared-GPU count (gpushare)
  volcano.sh/gpu-memory  — per-slice device memory MiB (gpushare slices)
  volcano.sh/vgpu-number / volcano.sh/vgpu-memory / volcano.sh/vgpu-cores
                         — vGPU slicing (memory MiB + core % per slice)

Node capacity comes from node annotations:
  volcano.sh/gpu-count   — physical cards
  volcano.sh/gpu-memory-per-card — MiB per card (288 GiB HBM3E on MI355X)

Dense resource dims cover the aggregate fit (the kernel's capacity
check); this pool covers *per-card packing* — two 160 GiB slices do not
fit one 288 GiB card even though 320 GiB aggregate might.
"""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List, Optional, Tuple

ANN_GPU_NUMBER = "volcano.sh/gpu-number"
ANN_GPU_MEMORY = "volcano.sh/gpu-memory"
ANN_VGPU_NUMBER = "volcano.sh/vgpu-number"
ANN_VGPU_MEMORY = "volcano.sh/vgpu-memory"
ANN_VGPU_CORES = "volcano.sh/vgpu-cores"
ANN_NODE_GPU_COUNT = "volcano.sh/gpu-count"
ANN_NODE_GPU_MEM = "volcano.sh/gpu-memory-per-card"
ANN_ASSIGNED = "volcano.sh/gpu-index"

MI355X_MEM_MIB = 288 * 1024     # 288 GB HBM3E per GPU


@dataclass
class DeviceRequest:
    count: int = 0          # slices wanted
    memory: int = 0         # MiB per slice (0 = exclusive whole card)
    cores: int = 0          # % of card compute per slice (vgpu)

    @property
    def exclusive(self) -> bool:
        return self.memory == 0 and self.cores == 0

    @classmethod
    def from_annotations(cls, ann: Dict[str, str]) -> Optional["DeviceRequest"]:
        if ANN_VGPU_NUMBER in ann:
            return cls(count=int(ann[ANN_VGPU_NUMBER]),
                       memory=int(ann.get(ANN_VGPU_MEMORY, 0)),
                       cores=int(ann.get(ANN_VGPU_CORES, 0)))
        if ANN_GPU_NUMBER in ann:
            return cls(count=int(ann[ANN_GPU_NUMBER]),
                       memory=int(ann.get(ANN_GPU_MEMORY, 0)))
        return None

    def signature(self) -> Tuple[int, int, int]:
        return (self.count, self.memory, self.cores)


@dataclass
class Card:
    index: int
    mem_total: int
    mem_used: int = 0
    cores_used: int = 0
    exclusive: bool = False
    slices: int = 0

    def fits(self, req: DeviceRequest) -> bool:
        if self.exclusive:
            return False
        if req.exclusive:
            return self.slices == 0
        if self.mem_used + req.memory > self.mem_total:
            return False
        if req.cores and self.cores_used + req.cores > 100:
            return False
        return True

    def take(self, req: DeviceRequest) -> None:
        if req.exclusive:
            self.exclusive = True
        else:
            self.mem_used += req.memory
            self.cores_used += req.cores
        self.slices += 1

    def release(self, req: DeviceRequest) -> None:
        if req.exclusive:
            self.exclusive = False
        else:
            self.mem_used = max(0, self.mem_used - req.memory)
            self.cores_used = max(0, self.cores_used - req.cores)
        self.slices = max(0, self.slices - 1)


class GPUDevicePool:
    """Best-fit slice packing over one node's cards (reference
    GPUDevice device_info.go:40-94)."""

    def __init__(self, node) -> None:
        ann = node.meta.annotations
        count = int(ann.get(ANN_NODE_GPU_COUNT, 0))
        mem = int(ann.get(ANN_NODE_GPU_MEM, MI355X_MEM_MIB))
        self.cards = [Card(i, mem) for i in range(count)]

    def fit(self, req: DeviceRequest) -> Optional[List[int]]:
        """Card indices able to take the request's slices (best-fit:
        fullest feasible card first), or None."""
        picked: List[int] = []
        # simulate on copies of the usage counters
        sims = [(c, c.mem_used, c.cores_used, c.slices, c.exclusive)
                for c in self.cards]
        for _ in range(req.count):
            best = None
            for k, (c, mu, cu, sl, ex) in enumerate(sims):
                if ex or (req.exclusive and sl > 0):
                    continue
                if not req.exclusive:
                    if mu + req.memory > c.mem_total:
                        continue
                    if req.cores and cu + req.cores > 100:
                        continue
                if best is None or mu > sims[best][1]:
                    best = k
            if best is None:
                return None
            c, mu, cu, sl, ex = sims[best]
            if req.exclusive:
                sims[best] = (c, mu, cu, sl + 1, True)
            else:
                sims[best] = (c, mu + req.memory, cu + req.cores, sl + 1, ex)
            picked.append(c.index)
        return picked

    def allocate(self, req: DeviceRequest) -> Optional[List[int]]:
        picked = self.fit(req)
        if picked is None:
            return None
        for idx in picked:
            self.cards[idx].take(req)
        return picked

    def release(self, req: DeviceRequest, indices: List[int]) -> None:
        for idx in indices:
            if 0 <= idx < len(self.cards):
                self.cards[idx].release(req)
