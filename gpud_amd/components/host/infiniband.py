"""infiniband — RDMA NIC port health (multi-node fabric NICs).

Reference: components/accelerator/nvidia/infiniband — a /sys/class/
infiniband parser (class/class.go:93-450: state/phys_state/rate/counters),
expected-port/rate thresholds (check_thresholds.go), and link-flap
detection with an auto-clear window (store/store.go). On MI355X clusters
the inter-node fabric is RoCE/IB NICs exactly as on the reference's
hardware, so this component carries over with the same mechanism; the
intra-node fabric is xGMI (accelerator-amd-xgmi).

The sysfs root is a constructor parameter so tests point it at fixture
trees (the reference does the same — class.go:93).
"""

from __future__ import annotations

import datetime
import os
from typing import Callable, Dict, List

from ...apiv1.types import (
    Event,
    EventType,
    HealthStateType,
    RepairActionType,
    SuggestedActions,
    utcnow,
)
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "infiniband"

SYSFS_ROOT = "/sys/class/infiniband"
FLAP_AUTO_CLEAR = datetime.timedelta(minutes=10)

# counters whose non-zero deltas indicate link problems
ERROR_COUNTERS = (
    "link_downed",
    "link_error_recovery",
    "symbol_error",
    "port_rcv_errors",
)


def _read(path: str) -> str:
    try:
        with open(path) as f:
            return f.read().strip()
    except OSError:
        return ""


def parse_ports(root: str = SYSFS_ROOT) -> List[Dict]:
    """All (device, port) entries with state/phys_state/rate/counters."""
    out: List[Dict] = []
    if not os.path.isdir(root):
        return out
    for dev in sorted(os.listdir(root)):
        ports_dir = os.path.join(root, dev, "ports")
        if not os.path.isdir(ports_dir):
            continue
        for port in sorted(os.listdir(ports_dir)):
            pdir = os.path.join(ports_dir, port)
            state = _read(os.path.join(pdir, "state"))  # "4: ACTIVE"
            phys = _read(os.path.join(pdir, "phys_state"))  # "5: LinkUp"
            rate = _read(os.path.join(pdir, "rate"))  # "400 Gb/sec (4X NDR)"
            try:
                rate_gbps = float(rate.split()[0]) if rate else 0.0
            except ValueError:
                rate_gbps = 0.0
            counters = {}
            cdir = os.path.join(pdir, "counters")
            for c in ERROR_COUNTERS:
                v = _read(os.path.join(cdir, c))
                counters[c] = int(v) if v.isdigit() else 0
            out.append(
                {
                    "device": dev,
                    "port": port,
                    "state": state,
                    "phys_state": phys,
                    "active": state.endswith("ACTIVE"),
                    "rate_gbps": rate_gbps,
                    "counters": counters,
                }
            )
    return out


class InfinibandComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        cfg = inst.config
        self.expected_ports = getattr(cfg, "expected_ib_ports", 0) if cfg else 0
        self.expected_rate_gbps = (
            getattr(cfg, "expected_ib_rate_gbps", 0.0) if cfg else 0.0
        )
        self.sysfs_root = SYSFS_ROOT
        self.get_ports: Callable[[], List[Dict]] = lambda: parse_ports(
            self.sysfs_root
        )
        self._last_counters: Dict[str, Dict[str, int]] = {}
        # SQLite ibports history (reference: infiniband/store/store.go —
        # drop/flap evaluation survives daemon restarts)
        self.link_store = None
        if inst.db_rw is not None and inst.db_ro is not None:
            try:
                from ...pkg.link_store import LinkStore

                self.link_store = LinkStore(
                    inst.db_rw, inst.db_ro, table_prefix="ibports"
                )
            except Exception:
                self.link_store = None
        self.flap_auto_clear_seconds = FLAP_AUTO_CLEAR.total_seconds()

    def can_set_healthy(self) -> bool:
        return True

    def set_healthy(self) -> None:
        if self.link_store is not None:
            self.link_store.set_tombstone()
        self._last_counters = {}

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME, "network", "fabric"]

    def is_supported(self) -> bool:
        return os.path.isdir(self.sysfs_root)

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    def check(self) -> CheckResult:
        ports = self.get_ports()
        if not ports:
            if self.expected_ports > 0:
                return CheckResult(
                    NAME,
                    health=HealthStateType.UNHEALTHY,
                    reason=f"expected {self.expected_ports} IB port(s), none found",
                    suggested_actions=SuggestedActions(
                        description="missing RDMA NICs",
                        repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                    ),
                )
            return CheckResult(NAME, reason="no InfiniBand/RoCE devices")
        active, down, slow = [], [], []
        extra = {}
        for p in ports:
            key = f"{p['device']}/{p['port']}"
            self._gauges.set(
                "infiniband_port_active", "1 when the port state is ACTIVE",
                1.0 if p["active"] else 0.0, device=p["device"], port=p["port"],
            )
            self._gauges.set(
                "infiniband_port_rate_gbps", "Port rate in Gb/s",
                p["rate_gbps"], device=p["device"], port=p["port"],
            )
            for c, v in p["counters"].items():
                self._gauges.set(
                    f"infiniband_{c}_total", f"IB counter {c}", v,
                    device=p["device"], port=p["port"],
                )
            extra[key] = f"{'up' if p['active'] else 'DOWN'} {p['rate_gbps']:g}Gb/s"
            if p["active"]:
                active.append(key)
                if (
                    self.expected_rate_gbps > 0
                    and p["rate_gbps"] < self.expected_rate_gbps
                ):
                    slow.append(key)
            else:
                down.append(key)
            # flap detection: rising link_downed counter
            prev = self._last_counters.get(key)
            cur = p["counters"]
            if (
                prev is not None
                and cur.get("link_downed", 0) > prev.get("link_downed", 0)
                and self._bucket is not None
            ):
                self._bucket.insert(
                    Event(
                        time=utcnow(),
                        component=NAME,
                        name="ib_port_flap",
                        type=EventType.CRITICAL,
                        message=f"IB port {key} link_downed "
                        f"{prev.get('link_downed', 0)}→{cur.get('link_downed', 0)}",
                    )
                )
            self._last_counters[key] = dict(cur)
        # persist the sweep; evaluate drops/flaps from the history store
        store_drops, store_flaps = [], []
        if self.link_store is not None and ports:
            try:
                self.link_store.insert(
                    [
                        {
                            "device": p["device"],
                            "port": int(str(p["port"]).strip() or 0),
                            "state": "active" if p["active"] else "down",
                            "rate_gb_sec": p["rate_gbps"] / 8.0,
                            "total_link_downed": p["counters"].get(
                                "link_downed", 0
                            ),
                        }
                        for p in ports
                    ]
                )
                ev = self.link_store.evaluate(
                    drop_sticky_window=10 * 60.0,
                    flap_auto_clear_window=self.flap_auto_clear_seconds,
                )
                store_drops, store_flaps = ev["drops"], ev["flaps"]
            except Exception:
                pass
        if store_drops:
            extra["port_drops"] = "; ".join(e.reason for e in store_drops)
        if store_flaps:
            extra["port_flaps"] = "; ".join(e.reason for e in store_flaps)
        if down or slow or (self.expected_ports > 0 and len(active) < self.expected_ports):
            parts = []
            if down:
                parts.append("ports down: " + ", ".join(down))
            if slow:
                parts.append(
                    f"ports below {self.expected_rate_gbps:g} Gb/s: " + ", ".join(slow)
                )
            if self.expected_ports > 0 and len(active) < self.expected_ports:
                parts.append(
                    f"only {len(active)}/{self.expected_ports} expected ports active"
                )
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="; ".join(parts),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="degraded RDMA fabric ports",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        if store_drops:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="IB port drop history: "
                + "; ".join(e.reason for e in store_drops),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="persistently down IB port",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        if store_flaps:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason="IB port flap history: "
                + "; ".join(e.reason for e in store_flaps),
                extra_info=extra,
            )
        # auto-clear window for recent flaps (reference store semantics)
        if self._bucket is not None:
            recent = self._bucket.find_by_name_since(
                "ib_port_flap", utcnow() - FLAP_AUTO_CLEAR
            )
            if recent:
                return CheckResult(
                    NAME,
                    health=HealthStateType.DEGRADED,
                    reason=f"IB port(s) flapped within the last "
                    f"{int(FLAP_AUTO_CLEAR.total_seconds() // 60)} min",
                    extra_info=extra,
                )
        return CheckResult(
            NAME,
            reason=f"{len(active)} IB port(s) active",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return InfinibandComponent(inst)
