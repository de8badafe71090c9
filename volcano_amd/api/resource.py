"""Resource vectors — the scheduler's unit of arithmetic.

Control-plane analog of the reference's ``pkg/scheduler/api/resource_info.go``
(Resource{MilliCPU, Memory, ScalarResources} + ~40 vector ops).  Unlike the
reference — which does per-field Go arithmetic on every comparison — the
MI355X design keeps *one canonical dense layout*: a resource is a fixed-order
float vector over the ``ResourceDims`` registry, so whole inventories pack
into ``[N, R]`` tensors that the HIP kernels consume directly.

The small ``Resource`` convenience class below is for control-plane code
(controllers, webhooks, tests); anything hot goes through tensors.
"""

from __future__ import annotations

import re
from typing import Dict, Iterable, List, Mapping, Optional

# Well-known resource names (mirror k8s / reference semantics).
CPU = "cpu"            # stored in millicores
MEMORY = "memory"      # stored in bytes
PODS = "pods"          # pod-count capacity
GPU = "amd.com/gpu"    # whole-GPU scalar resource (the reference's nvidia.com/gpu analog)

DEFAULT_DIMS = (CPU, MEMORY, PODS, GPU)

_MEM_SUFFIX = {
    "k": 1000, "M": 1000 ** 2, "G": 1000 ** 3, "T": 1000 ** 4, "P": 1000 ** 5,
    "Ki": 1024, "Mi": 1024 ** 2, "Gi": 1024 ** 3, "Ti": 1024 ** 4, "Pi": 1024 ** 5,
}

# A quantity that is smaller than this is considered zero for fit/compare
# purposes (the reference uses minResource = 0.1 in resource_info.go).
MIN_RESOURCE = 0.1


def parse_quantity(value, resource: str = "") -> float:
    """Parse a k8s-style quantity into canonical units.

    cpu → millicores ("2" → 2000, "500m" → 500); memory → bytes
    ("1Gi" → 1073741824); everything else → plain float count.
    """
    if isinstance(value, (int, float)):
        v = float(value)
        return v * 1000.0 if resource == CPU else v
    s = str(value).strip()
    if resource == CPU:
        if s.endswith("m"):
            return float(s[:-1])
        return float(s) * 1000.0
    m = re.fullmatch(r"([0-9.eE+-]+)([A-Za-z]*)", s)
    if not m:
        raise ValueError(f"unparseable quantity {value!r}")
    num, suffix = float(m.group(1)), m.group(2)
    if not suffix:
        return num
    if suffix in _MEM_SUFFIX:
        return num * _MEM_SUFFIX[suffix]
    if suffix == "m":  # milli of an extended resource
        return num / 1000.0
    raise ValueError(f"unknown quantity suffix {suffix!r} in {value!r}")


class ResourceDims:
    """Registry mapping resource names to dense tensor columns.

    The first ``len(DEFAULT_DIMS)`` columns are fixed; scalar resources seen
    in the inventory are appended.  Frozen once tensors are built — adding a
    dim later forces a repack (the snapshot layer handles that).
    """

    def __init__(self, names: Iterable[str] = DEFAULT_DIMS):
        self.names: List[str] = []
        self.index: Dict[str, int] = {}
        for n in names:
            self.add(n)

    def add(self, name: str) -> int:
        if name not in self.index:
            self.index[name] = len(self.names)
            self.names.append(name)
        return self.index[name]

    def __len__(self) -> int:
        return len(self.names)

    def __contains__(self, name: str) -> bool:
        return name in self.index

    def vector(self, mapping: Mapping[str, float], parse: bool = False) -> List[float]:
        """Dense vector for ``mapping`` (unknown dims are added)."""
        out = [0.0] * len(self.names)
        for k, v in mapping.items():
            i = self.add(k)
            if i >= len(out):
                out.extend([0.0] * (i + 1 - len(out)))
            out[i] = parse_quantity(v, k) if parse else float(v)
        if len(out) < len(self.names):
            out.extend([0.0] * (len(self.names) - len(out)))
        return out


def normalize_dra_keys(r: "Resource") -> "Resource":
    """Translate the reference's DRA quota keys (``count/<deviceClass>``
    — capacity.go DeviceClassCountPrefix) into the dense device-class
    dims (``dra:<class>``) so queue capability/deserved/guarantee rows
    enforce device-class quotas through the SAME in-kernel clamp as any
    other resource."""
    if not any(k.startswith("count/") for k in r.q):
        return r
    return Resource({(f"dra:{k[6:]}" if k.startswith("count/") else k): v
                     for k, v in r.q.items()})


class Resource:
    """Small dict-backed resource vector for control-plane code.

    Mirrors the operation set of the reference's ``Resource``
    (resource_info.go: Add/Sub/LessEqual/FitDelta/Diff/...), with canonical
    units (cpu in millicores, memory in bytes).
    """

    __slots__ = ("q",)

    def __init__(self, q: Optional[Mapping[str, float]] = None):
        self.q: Dict[str, float] = dict(q) if q else {}

    # -- construction -------------------------------------------------------
    @classmethod
    def from_spec(cls, spec: Optional[Mapping[str, object]]) -> "Resource":
        """Build from a k8s-style resource map ({"cpu": "500m", ...})."""
        r = cls()
        if spec:
            for k, v in spec.items():
                r.q[k] = r.q.get(k, 0.0) + parse_quantity(v, k)
        return r

    def clone(self) -> "Resource":
        return Resource(self.q)

    # -- accessors ----------------------------------------------------------
    def get(self, name: str) -> float:
        return self.q.get(name, 0.0)

    @property
    def milli_cpu(self) -> float:
        return self.get(CPU)

    @property
    def memory(self) -> float:
        return self.get(MEMORY)

    def is_empty(self) -> bool:
        return all(v < MIN_RESOURCE for v in self.q.values())

    def resource_names(self) -> List[str]:
        return list(self.q)

    # -- arithmetic (in-place, returning self, like the reference) ----------
    def add(self, other: "Resource") -> "Resource":
        for k, v in other.q.items():
            self.q[k] = self.q.get(k, 0.0) + v
        return self

    def sub(self, other: "Resource") -> "Resource":
        """Saturating subtract (reference sub() panics; we clamp at 0 like
        its SubWithoutAssert / saturating.go path)."""
        for k, v in other.q.items():
            self.q[k] = max(self.q.get(k, 0.0) - v, 0.0)
        return self

    def multi(self, ratio: float) -> "Resource":
        for k in self.q:
            self.q[k] *= ratio
        return self

    def set_max(self, other: "Resource") -> "Resource":
        for k, v in other.q.items():
            self.q[k] = max(self.q.get(k, 0.0), v)
        return self

    def min_dimension(self, other: "Resource") -> "Resource":
        """Per-dim min (reference MinDimensionResource)."""
        for k in list(self.q):
            self.q[k] = min(self.q[k], other.q.get(k, 0.0))
        return self

    # -- comparisons --------------------------------------------------------
    def less_equal(self, other: "Resource", zero_ok: bool = True) -> bool:
        """self <= other in every dimension self has (reference LessEqual
        with the `zero` defaulting strategy: dims absent from `other` count
        as 0)."""
        for k, v in self.q.items():
            if v < MIN_RESOURCE:
                continue
            if v > other.q.get(k, 0.0) + MIN_RESOURCE:
                return False
        return True

    def less_partly(self, other: "Resource") -> bool:
        """True if self < other in at least one dimension (reference
        LessPartly)."""
        return any(v < other.q.get(k, 0.0) - MIN_RESOURCE for k, v in self.q.items()) or any(
            k not in self.q and v > MIN_RESOURCE for k, v in other.q.items()
        )

    def fit_delta(self, req: "Resource") -> "Resource":
        """Remaining headroom after fitting req (may go conceptually
        negative; clamped report via diff)."""
        out = self.clone()
        for k, v in req.q.items():
            out.q[k] = out.q.get(k, 0.0) - v
        return out

    def diff(self, other: "Resource"):
        """(increased, decreased) vs other — reference Diff."""
        inc, dec = Resource(), Resource()
        for k in set(self.q) | set(other.q):
            d = self.q.get(k, 0.0) - other.q.get(k, 0.0)
            if d > MIN_RESOURCE:
                inc.q[k] = d
            elif d < -MIN_RESOURCE:
                dec.q[k] = -d
        return inc, dec

    def __eq__(self, other) -> bool:
        if not isinstance(other, Resource):
            return NotImplemented
        keys = set(self.q) | set(other.q)
        return all(abs(self.q.get(k, 0.0) - other.q.get(k, 0.0)) < MIN_RESOURCE for k in keys)

    def __repr__(self) -> str:
        body = ", ".join(f"{k}={v:g}" for k, v in sorted(self.q.items()))
        return f"Resource({body})"

    # -- dense interop ------------------------------------------------------
    def to_vector(self, dims: ResourceDims) -> List[float]:
        return dims.vector(self.q)

    @classmethod
    def from_vector(cls, vec, dims: ResourceDims) -> "Resource":
        return cls({n: float(vec[i]) for n, i in dims.index.items() if i < len(vec) and float(vec[i]) != 0.0})
