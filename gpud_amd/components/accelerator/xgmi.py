"""accelerator-amd-xgmi — inter-GPU xGMI fabric health.

The single-node MI355X replacement for the reference's nvlink +
fabric-manager + infiniband trio (reference:
components/accelerator/nvidia/nvlink/nvlink.go:86-93 per-link counters;
fabric-manager probes; infiniband port state/flap store — SURVEY.md §5
distributed-backend note): per-GPU link states (7 point-to-point links ×
≈153 GB/s on an 8-GPU node), per-link traffic counters from amdsmi link
metrics, the device xGMI error status, and an expected-link-count rule.

Link-flap detection: a link observed DOWN that was previously UP records an
event (the infiniband flap-store analog, simplified to the event bucket).
"""

from __future__ import annotations

import datetime
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
from .shared import SmiComponentMixin

NAME = "accelerator-amd-xgmi"

LINK_UP = 1
LINK_DOWN = 0

# flap auto-clear window (reference: infiniband store
# flap_auto_clear_window.go): a link that went down recently keeps the
# component Degraded for this long after recovery, so a flapping link
# cannot blink the node healthy between polls
FLAP_AUTO_CLEAR = datetime.timedelta(minutes=10)

# per-product expected xGMI link counts on a fully-populated node
# (reference: infiniband threshold_default.go per-product port defaults).
# MI3xx OAM modules expose 7 point-to-point xGMI links on an 8-GPU mesh;
# the default applies only when >1 GPU is visible (a single-GPU box
# legitimately reports no peer links).
PRODUCT_EXPECTED_LINKS = (
    ("MI355", 7),
    ("MI350", 7),
    ("MI325", 7),
    ("MI300", 7),
)


def expected_links_for_product(product: str, n_gpus: int) -> int:
    """Default expected up-link count by product name; 0 = no expectation."""
    if n_gpus < 2:
        return 0
    for marker, links in PRODUCT_EXPECTED_LINKS:
        if marker.lower() in (product or "").lower():
            # a partially populated node has fewer peers than links
            return min(links, n_gpus - 1)
    return 0


class XGMIComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._cfg = inst.config
        self.expected_links = (
            getattr(self._cfg, "expected_xgmi_link_count", 0)
            if self._cfg else 0
        )
        self._last_states: Dict[str, List[int]] = {}
        self._last_traffic: Dict = {}
        # SQLite link-state history: drop/flap evaluation survives daemon
        # restarts (reference: infiniband store/store.go:57-319)
        self.link_store = None
        if inst.db_rw is not None and inst.db_ro is not None:
            try:
                from ...pkg.link_store import LinkStore

                self.link_store = LinkStore(
                    inst.db_rw, inst.db_ro, table_prefix="xgmi"
                )
            except Exception:
                self.link_store = None
        self.flap_auto_clear_seconds = FLAP_AUTO_CLEAR.total_seconds()
        self.get_snapshots: Callable = (
            self._shared.get if self._shared is not None else lambda: {}
        )
        self.get_devices: Callable = (
            self._smi.devices if self._smi is not None else dict
        )

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", "fabric", NAME]

    def is_supported(self) -> bool:
        return self._smi is not None and self._smi.exists

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    def can_set_healthy(self) -> bool:
        return True

    def set_healthy(self) -> None:
        """Tombstone the link history: cleared drop/flap findings stay
        cleared across restarts (reference: store tombstone semantics).
        The event-bucket flap window is trimmed too so the auto-clear
        window does not resurface the cleared finding."""
        if self.link_store is not None:
            self.link_store.set_tombstone()
        if self._bucket is not None:
            try:
                import time as _time

                self._bucket.purge(int(_time.time()) + 1)
            except Exception:
                pass
        self._last_states = {}

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        snaps = self.get_snapshots()
        down_by_uuid: Dict[str, List[int]] = {}
        err_by_uuid: Dict[str, int] = {}
        missing_links: List[str] = []
        extra: Dict[str, str] = {}
        any_links = False
        # control-plane updateConfig mutates the shared Config live
        # (reference: SetDefault* setters re-read per check)
        expected = (
            getattr(self._cfg, "expected_xgmi_link_count", 0)
            if self._cfg else 0
        ) or self.expected_links
        if expected <= 0:
            expected = expected_links_for_product(
                getattr(self._smi, "product_name", "") or "", len(snaps)
            )
        store_rows: List[Dict] = []
        for uuid, snap in snaps.items():
            x = snap.get("xgmi_link_status")
            if x:
                any_links = True
                states = list(x.get("states") or [])
                up = sum(1 for s in states if s == LINK_UP)
                down = [i for i, s in enumerate(states) if s == LINK_DOWN]
                self._gauges.set(
                    "accelerator_amd_xgmi_links_up",
                    "xGMI links in UP state",
                    up,
                    uuid=uuid,
                )
                self._gauges.set(
                    "accelerator_amd_xgmi_links_total",
                    "Total xGMI links",
                    len(states),
                    uuid=uuid,
                )
                extra[f"{uuid}.links"] = f"{up}/{len(states)} up"
                if down:
                    down_by_uuid[uuid] = down
                # flap detection vs previous observation
                prev = self._last_states.get(uuid)
                if prev is not None and self._bucket is not None:
                    for i, s in enumerate(states):
                        if i < len(prev) and prev[i] == LINK_UP and s == LINK_DOWN:
                            self._bucket.insert(
                                Event(
                                    time=utcnow(),
                                    component=NAME,
                                    name="amd_xgmi_link_down",
                                    type=EventType.CRITICAL,
                                    message=f"xGMI link {i} on {uuid} went DOWN",
                                )
                            )
                self._last_states[uuid] = states
                store_rows.extend(
                    {
                        "device": uuid,
                        "port": i,
                        "state": "active" if s == LINK_UP else "down",
                    }
                    for i, s in enumerate(states)
                )
                if expected > 0 and up < expected:
                    missing_links.append(uuid)
            err = snap.get("xgmi_error_status")
            if err is not None and int(err) != 0:
                err_by_uuid[uuid] = int(err)
                self._gauges.set(
                    "accelerator_amd_xgmi_error_status",
                    "xGMI error status (0=no errors)",
                    int(err),
                    uuid=uuid,
                )
        # per-link traffic counters (separate, heavier SMI call) + derived
        # per-second rates from deltas between this component's polls
        import time as _time

        try:
            now_mono = _time.monotonic()
            for uuid, dev in self.get_devices().items():
                lm = dev.link_metrics()
                for i, link in enumerate(lm.get("links", [])):
                    if int(link.get("link_type", 0)) != 2:  # XGMI only
                        continue
                    read_kb = float(link.get("read_kb", 0))
                    write_kb = float(link.get("write_kb", 0))
                    self._gauges.set(
                        "accelerator_amd_xgmi_read_kb_total",
                        "Accumulated xGMI read traffic (KB)",
                        read_kb,
                        uuid=uuid,
                        link=str(i),
                    )
                    self._gauges.set(
                        "accelerator_amd_xgmi_write_kb_total",
                        "Accumulated xGMI write traffic (KB)",
                        write_kb,
                        uuid=uuid,
                        link=str(i),
                    )
                    key = (uuid, i)
                    prev = self._last_traffic.get(key)
                    if prev is not None:
                        dt = now_mono - prev[2]
                        if dt > 0:
                            self._gauges.set(
                                "accelerator_amd_xgmi_read_bytes_per_second",
                                "xGMI read rate between polls",
                                max(read_kb - prev[0], 0) * 1024.0 / dt,
                                uuid=uuid,
                                link=str(i),
                            )
                            self._gauges.set(
                                "accelerator_amd_xgmi_write_bytes_per_second",
                                "xGMI write rate between polls",
                                max(write_kb - prev[1], 0) * 1024.0 / dt,
                                uuid=uuid,
                                link=str(i),
                            )
                    self._last_traffic[key] = (read_kb, write_kb, now_mono)
        except Exception:
            pass  # traffic counters are best-effort

        # persist the sweep + evaluate drops/flaps from the SQLite history
        # (reference: infiniband store — restart-surviving flap detection)
        store_flaps = []
        store_drops = []
        if self.link_store is not None and store_rows:
            try:
                self.link_store.insert(store_rows)
                ev = self.link_store.evaluate(
                    drop_sticky_window=10 * 60.0,
                    flap_auto_clear_window=self.flap_auto_clear_seconds,
                )
                store_drops = ev["drops"]
                store_flaps = ev["flaps"]
            except Exception:
                pass
        if store_drops:
            extra["link_drops"] = "; ".join(e.reason for e in store_drops)
        if store_flaps:
            extra["link_flaps"] = "; ".join(e.reason for e in store_flaps)

        if down_by_uuid or err_by_uuid or missing_links:
            parts = []
            if down_by_uuid:
                parts.append(
                    "links down: "
                    + "; ".join(f"{u} {links}" for u, links in down_by_uuid.items())
                )
            if err_by_uuid:
                parts.append(
                    "xgmi errors on: " + ", ".join(err_by_uuid.keys())
                )
            if missing_links:
                parts.append(
                    f"fewer than {expected} links up on: "
                    + ", ".join(missing_links)
                )
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="; ".join(parts),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="degraded xGMI fabric",
                    repair_actions=[
                        RepairActionType.REBOOT_SYSTEM,
                        RepairActionType.HARDWARE_INSPECTION,
                    ],
                ),
            )
        # store-evaluated drops/flaps: a persistent drop that has not
        # stabilized, or a flapping link inside its auto-clear window,
        # keeps the component unhealthy/degraded even though the links are
        # up right now (reference: infiniband drop sticky window + flap
        # auto-clear window)
        if store_drops:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="xGMI link drop history: "
                + "; ".join(e.reason for e in store_drops),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="persistently down xGMI link",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        if store_flaps:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason="xGMI link flap history: "
                + "; ".join(e.reason for e in store_flaps),
                extra_info=extra,
            )
        # flap auto-clear: recent down events keep the state Degraded even
        # after the link recovered (reference infiniband flap store)
        if self._bucket is not None:
            recent_flaps = self._bucket.find_by_name_since(
                "amd_xgmi_link_down", utcnow() - FLAP_AUTO_CLEAR
            )
            if recent_flaps:
                return CheckResult(
                    NAME,
                    health=HealthStateType.DEGRADED,
                    reason=(
                        f"xGMI link(s) flapped within the last "
                        f"{int(FLAP_AUTO_CLEAR.total_seconds() // 60)} min "
                        f"({len(recent_flaps)} down event(s)); links currently up"
                    ),
                    extra_info=extra,
                )
        reason = (
            f"all xGMI links healthy on {len(snaps)} GPU(s)"
            if any_links
            else "no xGMI links reported (single GPU?)"
        )
        return CheckResult(NAME, reason=reason, extra_info=extra or None)


def new(inst: GPUdInstance) -> Component:
    return XGMIComponent(inst)
