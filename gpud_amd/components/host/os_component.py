"""os — kernel/uptime/reboots, zombie & D-state processes, kernel panics.

Reference: components/os (kernel/os version, uptime, reboot events from
RebootEventStore, zombie-count degraded/unhealthy thresholds, D-state
(uninterruptible-sleep) tracker, pstore kernel-panic scan, fd usage —
os/component.go:69-87,176).
"""

from __future__ import annotations

import datetime
from typing import Callable, List, Optional

import psutil

from ...apiv1.types import (
    Event,
    EventType,
    HealthStateType,
    RepairActionType,
    SuggestedActions,
)
from ...pkg import host as pkghost
from ...pkg.pstore import Scanner
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "os"

DEFAULT_ZOMBIE_DEGRADED = 1000
DEFAULT_ZOMBIE_UNHEALTHY = 2000


def file_nr() -> tuple:
    """(allocated, maximum) file handles from /proc/sys/fs/file-nr
    (reference: os component fd usage)."""
    try:
        with open("/proc/sys/fs/file-nr") as f:
            parts = f.read().split()
            return int(parts[0]), int(parts[2])
    except (OSError, IndexError, ValueError):
        return 0, 0


def count_process_states() -> dict:
    zombies = 0
    dstate = 0
    total = 0
    for p in psutil.process_iter(["status"]):
        total += 1
        st = p.info.get("status")
        if st == psutil.STATUS_ZOMBIE:
            zombies += 1
        elif st == psutil.STATUS_DISK_SLEEP:
            dstate += 1
    return {"total": total, "zombies": zombies, "dstate": dstate}


class OSComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._reboot_store = inst.reboot_event_store
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        cfg = inst.config
        self.zombie_degraded = (
            getattr(cfg, "zombie_degraded_threshold", DEFAULT_ZOMBIE_DEGRADED)
            if cfg
            else DEFAULT_ZOMBIE_DEGRADED
        )
        self.zombie_unhealthy = (
            getattr(cfg, "zombie_unhealthy_threshold", DEFAULT_ZOMBIE_UNHEALTHY)
            if cfg
            else DEFAULT_ZOMBIE_UNHEALTHY
        )
        self._pstore: Optional[Scanner] = None
        if inst.db_rw is not None and inst.db_ro is not None:
            try:
                self._pstore = Scanner(inst.db_rw, inst.db_ro)
            except Exception:
                self._pstore = None
        self.get_process_states: Callable = count_process_states

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def events(self, since: datetime.datetime) -> List[Event]:
        evs: List[Event] = []
        if self._reboot_store is not None:
            evs.extend(self._reboot_store.get_reboot_events(since))
        if self._bucket is not None:
            evs.extend(
                e for e in self._bucket.get(since) if e.name != "reboot"
            )
        evs.sort(key=lambda e: e.time, reverse=True)
        return evs

    def check(self) -> CheckResult:
        states = self.get_process_states()
        self._gauges.set("os_zombie_processes", "Zombie process count", states["zombies"])
        self._gauges.set(
            "os_dstate_processes",
            "Uninterruptible-sleep (D-state) process count",
            states["dstate"],
        )
        self._gauges.set(
            "os_uptime_seconds", "Seconds since boot", pkghost.uptime_seconds()
        )
        fd_alloc, fd_max = file_nr()
        if fd_max > 0:
            self._gauges.set(
                "os_file_handles_allocated", "System-wide allocated file handles",
                fd_alloc,
            )
            self._gauges.set(
                "os_file_handles_usage_percent",
                "Allocated file handles as percent of the system maximum",
                100.0 * fd_alloc / fd_max,
            )
        # pstore kernel-panic scan (new findings become Fatal events)
        panic_findings = []
        if self._pstore is not None:
            try:
                panic_findings = self._pstore.scan()
                for fname, sig, mtime in panic_findings:
                    if self._bucket is not None:
                        self._bucket.insert(
                            Event(
                                time=mtime,
                                component=NAME,
                                name="kernel_panic",
                                type=EventType.FATAL,
                                message=f"pstore {fname}: {sig}",
                            )
                        )
            except Exception:
                pass
        extra = {
            "kernel_version": pkghost.kernel_version(),
            "os_image": pkghost.os_image(),
            "boot_id": pkghost.boot_id(),
            "uptime_seconds": str(int(pkghost.uptime_seconds())),
            "zombies": str(states["zombies"]),
            "dstate": str(states["dstate"]),
        }
        if panic_findings:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"kernel panic records found in pstore: "
                + ", ".join(f[0] for f in panic_findings),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="kernel panic detected on a previous boot",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        if states["zombies"] >= self.zombie_unhealthy:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"{states['zombies']} zombie processes (>= {self.zombie_unhealthy})",
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="runaway zombie process accumulation",
                    repair_actions=[RepairActionType.REBOOT_SYSTEM],
                ),
            )
        if states["zombies"] >= self.zombie_degraded:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=f"{states['zombies']} zombie processes (>= {self.zombie_degraded})",
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"os healthy (kernel {pkghost.kernel_version()}, "
            f"uptime {int(pkghost.uptime_seconds())}s)",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return OSComponent(inst)
