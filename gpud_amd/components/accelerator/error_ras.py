"""accelerator-amd-error-ras — the flagship amdgpu error-event component.

The AMD-native equivalent of the reference's Xid component
(reference: components/accelerator/nvidia/xid/component.go — live
/dev/kmsg follow + full-ring re-read on Start, catalog lookup, health
state machine merging kmsg events with reboot events and SetHealthy
markers, reboot-count-aware escalation — xid/health_state.go:61-97,
set_healthy.go): amdgpu/KFD/RAS kernel messages are matched against the
curated catalog (gpud_amd/pkg/ras_catalog.py), persisted as events, and the
most recent *critical* event drives the health state:

  * first occurrence since the last reboot ⇒ suggested action from the
    catalog (typically REBOOT_SYSTEM);
  * the same event class recurring after ``reboot_threshold`` reboots ⇒
    escalated to HARDWARE_INSPECTION;
  * a reboot or a SetHealthy marker AFTER the event clears the state.
"""

from __future__ import annotations

import datetime
from typing import Callable, List, Optional

from ...apiv1.types import (
    Event,
    EventType,
    HealthStateType,
    RepairActionType,
    SuggestedActions,
    utcnow,
)
from ...pkg import ras_catalog
from ...pkg.kmsg.syncer import MatchResult, Syncer
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "accelerator-amd-error-ras"

SET_HEALTHY_EVENT = "SetHealthy"
DEFAULT_REBOOT_THRESHOLD = 2
LOOKBACK = datetime.timedelta(days=3)


def _match(line: str) -> Optional[MatchResult]:
    res = ras_catalog.match(line)
    if res is None:
        return None
    d, groups = res
    return MatchResult(
        name=d.name, event_type=d.event_type, message=line, extra_info=groups or None
    )


class ErrorRASComponent(TickerComponent):
    poll_interval = 60.0

    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._inst = inst
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._reboot_store = inst.reboot_event_store
        self._kmsg = inst.kmsg_reader
        self._syncer: Optional[Syncer] = None
        cfg = inst.config
        self.reboot_threshold = int(
            getattr(cfg, "ras_reboot_threshold", DEFAULT_REBOOT_THRESHOLD)
            or DEFAULT_REBOOT_THRESHOLD
        )
        # per-event-name overrides (reference: per-Xid thresholds,
        # xid/threshold.go + cmd/gpud/run/command.go xid-thresholds)
        self.event_thresholds = dict(
            getattr(cfg, "ras_event_thresholds", None) or {}
        )
        self.get_now: Callable = utcnow

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        return True  # kmsg matching works wherever /dev/kmsg is readable

    def start(self) -> None:
        # replay the kmsg ring (reference xid component.go:141-152 Start),
        # then follow live messages via the shared watcher
        if self._kmsg is not None and self._bucket is not None:
            self._syncer = Syncer(self._kmsg, _match, self._bucket)
            try:
                history = self._kmsg.read_all()
                self._syncer.replay(history)
            except Exception:
                pass
        super().start()

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    # -- SetHealthy (reference: xid/set_healthy.go) -------------------------

    def can_set_healthy(self) -> bool:
        return True

    def set_healthy(self) -> None:
        if self._bucket is not None:
            self._bucket.insert(
                Event(
                    time=self.get_now(),
                    component=NAME,
                    name=SET_HEALTHY_EVENT,
                    type=EventType.INFO,
                    message="health state manually cleared",
                )
            )
        self.trigger_check()

    # -- health state machine ----------------------------------------------

    def check(self) -> CheckResult:
        if self._bucket is None:
            return CheckResult(NAME, reason="no event store; kmsg matching disabled")
        now = self.get_now()
        since = now - LOOKBACK
        events = self._bucket.get(since)  # newest first

        # the newest critical catalog event, unless cleared by a later
        # reboot or SetHealthy marker
        last_clear: Optional[datetime.datetime] = None
        for ev in events:
            if ev.name == SET_HEALTHY_EVENT:
                last_clear = ev.time
                break
        reboots: List[Event] = (
            self._reboot_store.get_reboot_events(since)
            if self._reboot_store is not None
            else []
        )
        if reboots:
            latest_reboot = max(r.time for r in reboots)
            if last_clear is None or latest_reboot > last_clear:
                last_clear = latest_reboot

        active: Optional[Event] = None
        active_detail = None
        occurrences = 0
        for ev in events:
            if ev.name == SET_HEALTHY_EVENT:
                continue
            d = ras_catalog.lookup(ev.name)
            if d is None or not d.critical:
                continue
            occurrences += 1
            if active is None and (last_clear is None or ev.time > last_clear):
                active = ev
                active_detail = d

        n_events = len([e for e in events if e.name != SET_HEALTHY_EVENT])
        if active is None or active_detail is None:
            return CheckResult(
                NAME,
                reason=f"no active critical amdgpu/RAS errors ({n_events} recent events)",
            )

        # reboot-count-aware escalation (reference xid/health_state.go:61-97):
        # how many reboots happened since this event class first appeared —
        # if we already rebooted >= threshold times and it came back, escalate.
        first_occurrence = min(
            (e.time for e in events if e.name == active.name), default=active.time
        )
        reboots_since = sum(1 for r in reboots if r.time > first_occurrence)
        actions = list(active_detail.repair_actions)
        threshold = int(
            self.event_thresholds.get(active.name, self.reboot_threshold)
        )
        if reboots_since >= threshold:
            actions = [RepairActionType.HARDWARE_INSPECTION]
            desc = (
                f"{active_detail.description} (recurred after {reboots_since} "
                "reboot(s) — escalate to hardware inspection)"
            )
        else:
            desc = active_detail.description
        return CheckResult(
            NAME,
            health=HealthStateType.UNHEALTHY,
            reason=f"{active.name}: {active.message[:200]}",
            suggested_actions=SuggestedActions(
                description=desc, repair_actions=actions
            )
            if actions
            else None,
            extra_info={
                "event_name": active.name,
                "event_type": active.type,
                "occurrences_lookback": str(occurrences),
                "reboots_since_first": str(reboots_since),
            },
        )


def new(inst: GPUdInstance) -> Component:
    return ErrorRASComponent(inst)
