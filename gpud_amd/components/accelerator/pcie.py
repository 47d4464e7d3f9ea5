"""accelerator-amd-pcie — host-link health: width/speed + error counters.

Reference analog: the NVLink replay/recovery/CRC error counters
(components/accelerator/nvidia/nvlink/nvlink.go:86-93) and the pci
component's ACS concern, applied to the MI355X's PCIe Gen5 x16 host link:

- **down-training**: current width/speed below the slot maximum (bad
  seating, BIOS lane config, signal integrity) ⇒ Degraded with a
  hardware-inspection suggestion — bandwidth-starved DMA is a silent
  job-slowdown on an otherwise "healthy" node;
- **link errors**: replay / L0→recovery / NAK counter DELTAS between
  polls become events; sustained error rate (same ≥0.6 events/min over
  10 min windowed rule as hw-slowdown) ⇒ Degraded.
"""

from __future__ import annotations

import datetime
from typing import Callable, Dict

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

NAME = "accelerator-amd-pcie"

EVENT_NAME = "amd_gpu_pcie_link_errors"
WINDOW = datetime.timedelta(minutes=10)
THRESHOLD_PER_MIN = 0.6

_ERR_COUNTERS = (
    ("replay_count", "replay"),
    ("l0_to_recovery_count", "L0-to-recovery"),
    ("nak_sent_count", "NAK sent"),
    ("nak_received_count", "NAK received"),
)


class PCIeComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._last: Dict[str, Dict[str, int]] = {}
        self._supported = True
        self.get_now: Callable = utcnow

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        return self._smi is not None and self._smi.exists

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        if not self._supported:
            return CheckResult(
                NAME, reason="PCIe info not supported by this driver"
            )
        now = self.get_now()
        downtrained = []
        extra: Dict[str, str] = {}
        n = 0
        for uuid, dev in self._smi.devices().items():
            try:
                pi = dev.pcie_info()
            except Exception:
                self._supported = False
                return CheckResult(
                    NAME, reason="PCIe info not supported by this driver"
                )
            n += 1
            width, speed = int(pi.get("width", 0)), int(pi.get("speed_mts", 0))
            max_w = int(pi.get("max_width", 0))
            # the header documents max speed in GT/s but the live driver
            # returns MT/s (measured: 32000 on a Gen5 x16 MI355X, where the
            # doc'd unit would read 32) — normalize either convention
            max_s_mts = int(pi.get("max_speed_gts", 0))
            if 0 < max_s_mts < 1000:
                max_s_mts *= 1000
            self._gauges.set(
                "accelerator_amd_pcie_link_width", "Current PCIe width",
                width, uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_pcie_link_speed_mts",
                "Current PCIe speed (MT/s)", speed, uuid=uuid,
            )
            extra[f"{uuid}.pcie"] = f"x{width}@{speed}MT/s"
            if (max_w and width and width < max_w) or (
                max_s_mts and speed and speed < max_s_mts
            ):
                downtrained.append(
                    f"{uuid}: x{width}@{speed} (slot x{max_w}@{max_s_mts})"
                )
            prev = self._last.get(uuid, {})
            cur: Dict[str, int] = {}
            bumped = []
            for key, desc in _ERR_COUNTERS:
                val = int(pi.get(key, 0))
                cur[key] = val
                self._gauges.set(
                    f"accelerator_amd_pcie_{key}",
                    f"Total PCIe {desc} count", val, uuid=uuid,
                )
                if key in prev and val > prev[key]:
                    bumped.append(f"{desc} +{val - prev[key]}")
            self._last[uuid] = cur
            if bumped and self._bucket is not None:
                self._bucket.insert(
                    Event(
                        time=now,
                        component=NAME,
                        name=EVENT_NAME,
                        type=EventType.WARNING,
                        message=f"PCIe link errors on {uuid}: {', '.join(bumped)}",
                    )
                )
        if downtrained:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason="PCIe link down-trained — " + "; ".join(downtrained[:8]),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="host link below slot capability: check seating, "
                    "BIOS lane configuration, signal integrity",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        if self._bucket is not None:
            recent = self._bucket.find_by_name_since(EVENT_NAME, now - WINDOW)
            per_min = len(recent) / WINDOW.total_seconds() * 60.0
            if per_min >= THRESHOLD_PER_MIN:
                return CheckResult(
                    NAME,
                    health=HealthStateType.DEGRADED,
                    reason=(
                        f"sustained PCIe link errors: {per_min:.2f} events/min "
                        f"over the last {int(WINDOW.total_seconds() // 60)} min"
                    ),
                    extra_info=extra,
                    suggested_actions=SuggestedActions(
                        description="recurring PCIe replays/NAKs on the host link",
                        repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                    ),
                )
        return CheckResult(
            NAME, reason=f"PCIe link healthy on {n} GPU(s)", extra_info=extra
        )


def new(inst: GPUdInstance) -> Component:
    return PCIeComponent(inst)
