"""accelerator-amd-cper — structured RAS records from the driver's CPER cache.

Complements accelerator-amd-error-ras (dmesg text matching — the reference
xid component analog, components/accelerator/nvidia/xid/component.go:141):
the amdgpu driver also caches binary CPER (Common Platform Error Record)
entries with a parsed severity, a notify-type GUID (MCE/CMC/PCIe/BOOT...)
and a unique record id — machine-readable RAS that survives dmesg ring
wraparound and needs no regex curation. Cursor-based drain:

- ``Start`` drains from cursor 0, replaying the driver's cache — the same
  "rebuild health from the ring on boot" semantic the reference xid
  component gets from re-reading /dev/kmsg (xid/component.go:581).
- each ``Check`` continues from the saved cursor; only NEW records insert
  events (dedup by record id, persisted in the event messages so daemon
  restarts do not re-alert).

Health: fatal record in the lookback window ⇒ Unhealthy (HW inspection);
non-fatal uncorrected ⇒ Degraded (reboot suggested); corrected ⇒ event
only. SetHealthy clears (component deregisters events via the bucket
purge like error-ras).
"""

from __future__ import annotations

import datetime
import re
import threading
from typing import Callable, Dict, List, Set

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

NAME = "accelerator-amd-cper"

EVENT_NAME = "amd_gpu_cper"
LOOKBACK = datetime.timedelta(hours=1)
_RID_RE = re.compile(r"record_id=([^\s,]+)")

# severity values from the amdsmi CPER header contract
SEV_NONFATAL_UNCORRECTED = 0
SEV_FATAL = 1
SEV_NONFATAL_CORRECTED = 2


class CPERComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._lock = threading.Lock()
        self._cursors: Dict[str, int] = {}
        self._seen: Set[str] = set()
        self._supported: bool = True
        self._counts: Dict[str, Dict[int, int]] = {}
        self.get_now: Callable = utcnow

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        return self._smi is not None and self._smi.exists

    def start(self) -> None:
        # restart-dedup: re-learn record ids already alerted on from the
        # durable event store before the first driver-cache replay
        if self._bucket is not None:
            for ev in self._bucket.get(self.get_now() - datetime.timedelta(days=7)):
                m = _RID_RE.search(ev.message or "")
                if m:
                    self._seen.add(m.group(1))
        super().start()

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    def set_healthy(self) -> None:
        # trim every stored CPER event (timestamp < now+1 covers the events
        # inserted this very second; purge uses a strict < comparison)
        if self._bucket is not None:
            self._bucket.purge(int(self.get_now().timestamp()) + 1)

    def _drain(self, uuid: str, dev) -> List[Dict]:
        with self._lock:
            cursor = self._cursors.get(uuid, 0)
        res = dev.cper_entries(cursor=cursor)
        if not res.get("supported", False):
            self._supported = False
            return []
        with self._lock:
            self._cursors[uuid] = int(res.get("cursor", cursor))
        return list(res.get("entries", []))

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        # once the driver reports the interface unsupported, stop paying a
        # per-GPU probe every cycle (support cannot appear at runtime)
        if not self._supported:
            return CheckResult(
                NAME, reason="CPER interface not supported by this driver"
            )
        now = self.get_now()
        fresh_by_uuid: Dict[str, List[Dict]] = {}
        for uuid, dev in self._smi.devices().items():
            try:
                entries = self._drain(uuid, dev)
            except Exception as e:  # noqa: BLE001 — one bad GPU must not stop the sweep
                return CheckResult(
                    NAME,
                    health=HealthStateType.UNHEALTHY,
                    reason=f"CPER drain failed on {uuid}",
                    error=str(e),
                )
            fresh = []
            counts = self._counts.setdefault(
                uuid,
                {SEV_NONFATAL_UNCORRECTED: 0, SEV_FATAL: 0, SEV_NONFATAL_CORRECTED: 0},
            )
            for e in entries:
                rid = str(e.get("record_id", "") or "")
                key = f"{uuid}:{rid}" if rid else ""
                if key and key in self._seen:
                    continue
                if key:
                    self._seen.add(key)
                sev = int(e.get("severity", SEV_NONFATAL_CORRECTED))
                counts[sev] = counts.get(sev, 0) + 1
                fresh.append(e)
                if self._bucket is not None:
                    etype = (
                        EventType.FATAL
                        if sev == SEV_FATAL
                        else EventType.WARNING
                        if sev == SEV_NONFATAL_UNCORRECTED
                        else EventType.INFO
                    )
                    nt = e.get("notify_type", "?")
                    # FRU-level attribution from the decoded section
                    # descriptors (UEFI CPER appendix N): FRU text names
                    # the failing replaceable unit, section type names the
                    # error class
                    sec_bits = []
                    for s in e.get("sections", []) or []:
                        label = s.get("type_name") or (
                            (s.get("type_guid") or "?")[:8]
                        )
                        fru = s.get("fru_text") or s.get("fru_id") or ""
                        sec_bits.append(
                            f"{label}" + (f" fru={fru}" if fru else "")
                        )
                    sec_txt = (
                        " [" + "; ".join(sec_bits) + "]" if sec_bits else ""
                    )
                    self._bucket.insert(
                        Event(
                            time=now,
                            component=NAME,
                            name=EVENT_NAME,
                            type=etype,
                            message=(
                                f"CPER {e.get('severity_name', sev)} record on "
                                f"{uuid}: notify={nt} "
                                f"record_id={uuid}:{rid or '?'} "
                                f"sections={e.get('section_count', 0)}"
                                f"{sec_txt}"
                            ),
                        )
                    )
            if fresh:
                fresh_by_uuid[uuid] = fresh
            for sev, label in (
                (SEV_FATAL, "fatal"),
                (SEV_NONFATAL_UNCORRECTED, "nonfatal_uncorrected"),
                (SEV_NONFATAL_CORRECTED, "corrected"),
            ):
                self._gauges.set(
                    f"accelerator_amd_cper_{label}_total",
                    f"CPER records of severity {label} since daemon start",
                    float(counts.get(sev, 0)),
                    uuid=uuid,
                )
        if not self._supported:
            return CheckResult(
                NAME, reason="CPER interface not supported by this driver"
            )
        # health from the recent window of stored events
        if self._bucket is not None:
            recent = self._bucket.find_by_name_since(EVENT_NAME, now - LOOKBACK)
            fatal = [e for e in recent if e.type == EventType.FATAL]
            uncorrected = [e for e in recent if e.type == EventType.WARNING]
            if fatal:
                return CheckResult(
                    NAME,
                    health=HealthStateType.UNHEALTHY,
                    reason=f"{len(fatal)} fatal CPER record(s) in the last "
                    f"{int(LOOKBACK.total_seconds() // 60)} min: "
                    f"{fatal[0].message}",
                    suggested_actions=SuggestedActions(
                        description="fatal RAS record reported by the GPU",
                        repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                    ),
                )
            if uncorrected:
                return CheckResult(
                    NAME,
                    health=HealthStateType.DEGRADED,
                    reason=f"{len(uncorrected)} uncorrected (non-fatal) CPER "
                    f"record(s) in the last "
                    f"{int(LOOKBACK.total_seconds() // 60)} min",
                    suggested_actions=SuggestedActions(
                        description="uncorrected RAS records; reboot to let bad-page "
                        "retirement run",
                        repair_actions=[RepairActionType.REBOOT_SYSTEM],
                    ),
                )
        n_new = sum(len(v) for v in fresh_by_uuid.values())
        return CheckResult(
            NAME,
            reason=(
                f"{n_new} new CPER record(s), none actionable"
                if n_new
                else "no CPER records"
            ),
        )


def new(inst: GPUdInstance) -> Component:
    return CPERComponent(inst)
