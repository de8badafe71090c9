"""Network QoS enforcement layer (reference ``pkg/networkqos``):
online/offline bandwidth watermarks enforced by an eBPF program attached
with ``tc filter ... direct-action`` (tc_linux.go:116-123) and driven
through eBPF map updates (utils/ebpf/map.go, throttling.go).

The enforcement BACKEND is abstracted: production would shell out to
``tc``/bpftool against a real interface; tests (and this clusterless
environment) use a recording backend that captures the exact command
sequence and map writes, so the control logic — watermark computation,
throttle adjustment loop — is exercised against the same contract."""

from __future__ import annotations

from dataclasses import dataclass
from typing import Dict, List

# eBPF map keys (reference utils/ebpf throttling config map)
KEY_ONLINE_BW_WATERMARK = "online-bandwidth-watermark"
KEY_OFFLINE_LOW = "offline-low-bandwidth"
KEY_OFFLINE_HIGH = "offline-high-bandwidth"
KEY_CHECK_INTERVAL = "check-interval"


class RecordingTcBackend:
    """Captures the tc command sequence + eBPF map state."""

    def __init__(self):
        self.commands: List[str] = []
        self.maps: Dict[str, int] = {}

    def run(self, cmd: str) -> None:
        self.commands.append(cmd)

    def map_update(self, key: str, value: int) -> None:
        self.maps[key] = value


@dataclass
class NetQoSConfig:
    """Online/offline watermarks in bps (reference cni conf)."""

    interface: str = "eth0"
    total_bps: int = 10_000_000_000
    online_watermark_pct: float = 0.8      # online traffic protection
    offline_low_pct: float = 0.1           # offline floor
    offline_high_pct: float = 0.4          # offline ceiling
    check_interval_ms: int = 100
    bpf_object: str = "/usr/share/bwmcli/bwm_tc.o"


class NetQoSEnforcer:
    """Attach + program + adjust (the CNI plugin and throttling loop)."""

    def __init__(self, config: NetQoSConfig, backend=None):
        self.config = config
        self.backend = backend if backend is not None else \
            RecordingTcBackend()
        self.attached = False

    def attach(self) -> None:
        """tc qdisc + filter attach (tc_linux.go:116-123)."""
        c, b = self.config, self.backend
        b.run(f"tc qdisc add dev {c.interface} clsact")
        b.run(f"tc filter add dev {c.interface} egress bpf direct-action "
              f"obj {c.bpf_object} sec tc")
        self.program_watermarks()
        self.attached = True

    def program_watermarks(self) -> None:
        b = self.backend
        c = self.config
        b.map_update(KEY_ONLINE_BW_WATERMARK,
                     int(c.total_bps * c.online_watermark_pct))
        b.map_update(KEY_OFFLINE_LOW, int(c.total_bps * c.offline_low_pct))
        b.map_update(KEY_OFFLINE_HIGH, int(c.total_bps * c.offline_high_pct))
        b.map_update(KEY_CHECK_INTERVAL, c.check_interval_ms)

    def adjust(self, online_bps_used: int) -> int:
        """Throttling loop (throttling.go): offline bandwidth floats
        between low and high watermarks depending on online pressure.
        Returns the offline limit programmed."""
        c, b = self.config, self.backend
        online_wm = int(c.total_bps * c.online_watermark_pct)
        lo = int(c.total_bps * c.offline_low_pct)
        hi = int(c.total_bps * c.offline_high_pct)
        if online_bps_used >= online_wm:
            limit = lo                       # online saturated: floor
        else:
            headroom = online_wm - online_bps_used
            limit = min(hi, lo + headroom)
        b.map_update(KEY_OFFLINE_HIGH, limit)
        return limit

    def detach(self) -> None:
        c, b = self.config, self.backend
        b.run(f"tc filter del dev {c.interface} egress")
        b.run(f"tc qdisc del dev {c.interface} clsact")
        self.attached = False
