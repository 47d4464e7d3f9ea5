"""accelerator-amd-throttle — HW slowdown / throttle events.

Reference: components/accelerator/nvidia/hw-slowdown (clock-event-reasons
bitmask, unhealthy when event frequency ≥0.6/min over a 10-minute window —
hw-slowdown/component.go:29-35). AMD-first: amdsmi violation status — the
accumulated throttler residencies (PROCHOT / PPT power / socket thermal /
VR thermal / HBM thermal) and their ``active_*`` flags. Each check where a
throttler is active records an event; the same ≥0.6 events/min-over-10-min
window drives the Unhealthy transition.
"""

from __future__ import annotations

import datetime
from typing import Callable, Dict, List, Tuple

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

NAME = "accelerator-amd-throttle"

WINDOW = datetime.timedelta(minutes=10)
THRESHOLD_PER_MIN = 0.6  # same rule as the reference
EVENT_NAME = "amd_gpu_throttle"

_THROTTLERS = (
    ("prochot_thrm", "PROCHOT (thermal)"),
    ("ppt_pwr", "package power (PPT)"),
    ("socket_thrm", "socket thermal"),
    ("vr_thrm", "voltage-regulator thermal"),
    ("hbm_thrm", "HBM thermal"),
)


class ThrottleComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self.get_snapshots: Callable = (
            self._shared.get if self._shared is not None else lambda: {}
        )
        self.get_now: Callable = utcnow
        # last residency accumulators per uuid: activity = positive delta
        # between this component's consecutive polls (the fast-path snapshot
        # reads gpu_metrics accumulators; amdsmi's own violation_status call
        # blocks ~100 ms/GPU double-sampling, so we difference ourselves)
        self._last_acc: Dict[str, Dict[str, int]] = {}

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        return self._smi is not None and self._smi.exists

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    def _active_throttlers(self, uuid: str, v: Dict) -> List[Tuple[str, str]]:
        out = []
        prev = self._last_acc.get(uuid, {})
        cur: Dict[str, int] = {"_counter": int(v.get("acc_counter", 0))}
        d_counter = cur["_counter"] - prev.get("_counter", 0)
        for key, desc in _THROTTLERS:
            acc = int(v.get(f"acc_{key}", 0))
            cur[key] = acc
            explicit = int(v.get(f"active_{key}", 0))
            delta = acc - prev.get(key, acc)
            rising = key in prev and delta > 0
            if explicit or rising:
                out.append((key, desc))
            # violation percentage over the window between polls
            # (PVIOL/TVIOL per the amdsmi gpu_metrics accumulator contract:
            # delta residency / delta accumulation counter * 100)
            if key in prev and d_counter > 0:
                self._gauges.set(
                    f"accelerator_amd_throttle_{key}_violation_percent",
                    f"{key} residency percent between polls (PVIOL/TVIOL-style)",
                    100.0 * max(delta, 0) / d_counter,
                    uuid=uuid,
                )
        self._last_acc[uuid] = cur
        return out

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        snaps = self.get_snapshots()
        now = self.get_now()
        active_by_uuid: Dict[str, List[str]] = {}
        extra = {}
        for uuid, snap in snaps.items():
            v = snap.get("violation")
            if not v:
                continue
            for key, _desc in _THROTTLERS:
                self._gauges.set(
                    f"accelerator_amd_throttle_acc_{key}",
                    f"Accumulated {key} throttler residency counter",
                    float(v.get(f"acc_{key}", 0)),
                    uuid=uuid,
                )
            active = self._active_throttlers(uuid, v)
            # raw SMU bitmask exported for operators (informational only —
            # see the NOTE below for why it is not a health signal)
            gm = snap.get("gpu_metrics") or {}
            self._gauges.set(
                "accelerator_amd_throttle_indep_status_bits",
                "Raw gpu_metrics indep_throttle_status bitmask",
                float(int(gm.get("indep_throttle_status", 0) or 0)),
                uuid=uuid,
            )
            # NOTE: gpu_metrics' indep_throttle_status bitmask is NOT used
            # as an activity signal: it carries benign always-set bits
            # (low-utilization / gfx-clk-below-host-limit) on idle GPUs —
            # measured: treating it as active inserted an event every poll,
            # grew the event table unboundedly (bench p50 2.0 -> 4.0 ms
            # over 2000 cycles) and would flag idle boards as throttling.
            # The named residency accumulators above are the real signal.
            if active:
                descs = [d for _k, d in active]
                active_by_uuid[uuid] = descs
                extra[f"{uuid}.throttle"] = ",".join(k for k, _d in active)
                if self._bucket is not None:
                    self._bucket.insert(
                        Event(
                            time=now,
                            component=NAME,
                            name=EVENT_NAME,
                            type=EventType.WARNING,
                            message=f"GPU {uuid} throttled: {', '.join(descs)}",
                        )
                    )
        # windowed frequency rule (reference hw-slowdown component.go:29-35)
        if self._bucket is not None:
            window_events = self._bucket.find_by_name_since(EVENT_NAME, now - WINDOW)
            per_min = len(window_events) / WINDOW.total_seconds() * 60.0
            self._gauges.set(
                "accelerator_amd_throttle_events_per_minute",
                "Throttle events per minute over the 10-minute window",
                per_min,
            )
            if per_min >= THRESHOLD_PER_MIN:
                return CheckResult(
                    NAME,
                    health=HealthStateType.UNHEALTHY,
                    reason=(
                        f"sustained throttling: {per_min:.2f} events/min over the "
                        f"last {int(WINDOW.total_seconds()//60)} min"
                    ),
                    extra_info=extra,
                    suggested_actions=SuggestedActions(
                        description="GPU thermally or power limited for a sustained period",
                        repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                    ),
                )
        if active_by_uuid:
            descs = "; ".join(f"{u}: {', '.join(d)}" for u, d in active_by_uuid.items())
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=f"throttling active — {descs}",
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"no active throttling on {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return ThrottleComponent(inst)
