"""os — kernel/uptime/reboots, zombie & D-state processes, kernel panics.

Reference: components/os (kernel/os version, uptime, reboot events from
RebootEventStore, zombie-count degraded/unhealthy thresholds, D-state
(uninterruptible-sleep) persistence tracker, pstore kernel-panic scan,
fd usage — os/component.go:69-87,176 + os/blocked_processes.go).

D-state tracking mechanism (reference os/blocked_processes.go): a D-state
task waits inside a kernel syscall and ignores even SIGKILL, but most
blocking disk I/O passes through D briefly — so flagging requires
PERSISTENCE across consecutive one-minute checks (default 5), with a
wall-time gate so trigger-check bursts cannot inflate the counter, and an
absence grace of one check so PID churn between enumeration and metadata
reads does not reset the counter. Escalation is name-gated: every
persistent D-state process degrades the component, but only names matching
the escalation regexes (default ^amd/^rocm management processes — the
reference gates on ^nvidia) mark it Unhealthy with a reboot suggestion,
and repeated reboots without recovery escalate to hardware inspection.
"""

from __future__ import annotations

import datetime
import re
import threading
import time
from dataclasses import dataclass, field
from typing import Callable, Dict, List, Optional, Tuple

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

# D-state persistence (reference: os/threshold.go + blocked_processes.go)
DEFAULT_DSTATE_PERSISTENCE_THRESHOLD = 5  # consecutive one-minute checks
DEFAULT_DSTATE_NAME_REGEXES = ["^amd", "^rocm"]  # management-process gate
DSTATE_ABSENCE_GRACE = 1  # checks a tracked PID may vanish (PID churn)
DSTATE_CHECK_INTERVAL_SECONDS = 60.0  # the cadence persistence calibrates to
DSTATE_REBOOT_THRESHOLD = 2  # reboots before escalating to HW inspection
MAX_DSTATE_OUTPUT = 20  # bounded payload


@dataclass
class BlockedProcess:
    pid: int
    name: str
    first_seen: float
    last_seen: float
    consecutive_checks: int

    @property
    def blocked_seconds(self) -> int:
        return int(self.last_seen - self.first_seen)


@dataclass
class _BlockedEntry:
    name: str
    first_seen: float
    last_seen: float
    consecutive_checks: int = 1
    absent_checks: int = 0


class BlockedProcessTracker:
    """Tracks D-state PIDs across checks (reference:
    os/blocked_processes.go blockedProcessTracker.update). A PID absent for
    at most DSTATE_ABSENCE_GRACE consecutive checks keeps its persistence
    counter (not extended); longer absence drops it (recovered). The
    wall-time gate (>= (threshold-1) check intervals) makes "N consecutive
    one-minute checks" robust against out-of-band trigger-check bursts."""

    def __init__(self) -> None:
        self._mu = threading.Lock()
        self._entries: Dict[int, _BlockedEntry] = {}

    def reset(self) -> None:
        """Operator set-healthy: a still-blocked process must re-earn the
        persistence threshold before being flagged again."""
        with self._mu:
            self._entries = {}

    def update(
        self,
        now: float,
        blocked: List[Tuple[int, str]],
        persistence_threshold: int = DEFAULT_DSTATE_PERSISTENCE_THRESHOLD,
    ) -> Dict[str, List[BlockedProcess]]:
        if persistence_threshold <= 0:
            persistence_threshold = DEFAULT_DSTATE_PERSISTENCE_THRESHOLD
        with self._mu:
            seen = set()
            first_seen: List[BlockedProcess] = []
            for pid, name in blocked:
                seen.add(pid)
                e = self._entries.get(pid)
                if e is None:
                    self._entries[pid] = _BlockedEntry(
                        name=name, first_seen=now, last_seen=now
                    )
                    first_seen.append(
                        BlockedProcess(pid, name, now, now, 1)
                    )
                    continue
                if name:
                    e.name = name
                e.last_seen = now
                e.consecutive_checks += 1
                e.absent_checks = 0
            cleared: List[BlockedProcess] = []
            for pid in list(self._entries):
                if pid in seen:
                    continue
                e = self._entries[pid]
                e.absent_checks += 1
                if e.absent_checks > DSTATE_ABSENCE_GRACE:
                    cleared.append(
                        BlockedProcess(
                            pid, e.name, e.first_seen, e.last_seen,
                            e.consecutive_checks,
                        )
                    )
                    del self._entries[pid]
            persistent: List[BlockedProcess] = []
            min_wall = (persistence_threshold - 1) * DSTATE_CHECK_INTERVAL_SECONDS
            for pid, e in sorted(self._entries.items()):
                if (
                    e.consecutive_checks >= persistence_threshold
                    and (e.last_seen - e.first_seen) >= min_wall
                ):
                    persistent.append(
                        BlockedProcess(
                            pid, e.name, e.first_seen, e.last_seen,
                            e.consecutive_checks,
                        )
                    )
            return {
                "persistent": persistent[:MAX_DSTATE_OUTPUT],
                "persistent_count": len(persistent),  # type: ignore[dict-item]
                "first_seen": first_seen,
                "cleared": cleared,
            }


# FD / PID pressure thresholds (reference: os/component.go:51-54 —
# defaultMaxAllocatedFileHandlesPctDegraded/Unhealthy, same for PIDs)
FD_USAGE_PCT_DEGRADED = 90.0
FD_USAGE_PCT_UNHEALTHY = 95.0
PID_USAGE_PCT_DEGRADED = 90.0
PID_USAGE_PCT_UNHEALTHY = 95.0


def pid_max() -> int:
    try:
        with open("/proc/sys/kernel/pid_max") as f:
            return int(f.read().strip())
    except (OSError, ValueError):
        return 0


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
    total = 0
    blocked: List[Tuple[int, str]] = []
    for p in psutil.process_iter(["status", "name", "pid"]):
        total += 1
        st = p.info.get("status")
        if st == psutil.STATUS_ZOMBIE:
            zombies += 1
        elif st == psutil.STATUS_DISK_SLEEP:
            blocked.append((p.info.get("pid") or 0, p.info.get("name") or ""))
    return {
        "total": total,
        "zombies": zombies,
        "dstate": len(blocked),
        "blocked": blocked,
    }


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
        # D-state persistence tracking (reference os/blocked_processes.go);
        # thresholds read from config on every check so updateConfig takes
        # effect on the running daemon (the reference's getThresholdsFunc
        # pattern). An empty regex set disables escalation entirely.
        self._blocked_tracker = BlockedProcessTracker()
        self.dstate_persistence_threshold = (
            getattr(cfg, "dstate_persistence_threshold",
                    DEFAULT_DSTATE_PERSISTENCE_THRESHOLD)
            if cfg else DEFAULT_DSTATE_PERSISTENCE_THRESHOLD
        )
        regexes = (
            getattr(cfg, "dstate_name_regexes", None)
            if cfg else None
        )
        self.dstate_name_regexes: List[str] = (
            list(regexes) if regexes is not None
            else list(DEFAULT_DSTATE_NAME_REGEXES)
        )
        self.get_time_now: Callable[[], float] = time.time
        self.get_file_nr: Callable = file_nr
        self.get_pid_max: Callable = pid_max

    def _dstate_name_matches(self, name: str) -> bool:
        return any(re.search(rx, name) for rx in self.dstate_name_regexes)

    def can_set_healthy(self) -> bool:
        return True

    def set_healthy(self) -> None:
        # a still-blocked process must re-earn the persistence threshold
        self._blocked_tracker.reset()

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
        fd_alloc, fd_max = self.get_file_nr()
        fd_pct = 0.0
        if fd_max > 0:
            fd_pct = 100.0 * fd_alloc / fd_max
            self._gauges.set(
                "os_file_handles_allocated", "System-wide allocated file handles",
                fd_alloc,
            )
            self._gauges.set(
                "os_file_handles_usage_percent",
                "Allocated file handles as percent of the system maximum",
                fd_pct,
            )
        pmax = self.get_pid_max()
        pid_pct = 0.0
        if pmax > 0:
            pid_pct = 100.0 * states["total"] / pmax
            self._gauges.set(
                "os_running_pids_usage_percent",
                "Running PIDs as percent of kernel.pid_max",
                pid_pct,
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
        # D-state persistence pass (reference: evaluateBlockedProcesses)
        now = self.get_time_now()
        upd = self._blocked_tracker.update(
            now,
            states.get("blocked", []),
            self.dstate_persistence_threshold,
        )
        persistent = upd["persistent"]
        self._gauges.set(
            "os_dstate_persistent_processes",
            "Processes blocked for >= the persistence threshold of checks",
            len(persistent),
        )

        extra = {
            "kernel_version": pkghost.kernel_version(),
            "os_image": pkghost.os_image(),
            "boot_id": pkghost.boot_id(),
            "uptime_seconds": str(int(pkghost.uptime_seconds())),
            "zombies": str(states["zombies"]),
            "dstate": str(states["dstate"]),
        }
        if persistent:
            extra["dstate_persistent"] = ", ".join(
                f"{b.name or '?'}[{b.pid}] blocked {b.blocked_seconds}s "
                f"({b.consecutive_checks} checks)"
                for b in persistent
            )
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
        escalating = [
            b for b in persistent
            if self.dstate_name_regexes and self._dstate_name_matches(b.name)
        ]
        if escalating:
            # name-gated escalation: reboot suggested; repeated reboots
            # without recovery escalate to hardware inspection (reference:
            # DefaultBlockedProcessRebootThreshold = 2)
            action = RepairActionType.REBOOT_SYSTEM
            desc = ("reboot the system to clear persistent D-state processes "
                    "stuck in uninterruptible kernel waits")
            if self._reboot_store is not None:
                first = min(b.first_seen for b in escalating)
                since = datetime.datetime.fromtimestamp(
                    first, tz=datetime.timezone.utc
                ) - datetime.timedelta(days=7)
                try:
                    reboots = self._reboot_store.reboot_count_since(since)
                except Exception:
                    reboots = 0
                if reboots >= DSTATE_REBOOT_THRESHOLD:
                    action = RepairActionType.HARDWARE_INSPECTION
                    desc = ("persistent D-state processes survive reboots; "
                            "inspect storage/driver hardware")
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="persistent D-state (uninterruptible sleep) processes "
                f"detected (persistent: {len(persistent)}): "
                + ", ".join(f"{b.name}[{b.pid}]" for b in escalating),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description=desc, repair_actions=[action]
                ),
            )
        if persistent:
            # persistent D-state outside the escalation gate: degraded —
            # D-state alone does not identify the cause (a dd on a wedged
            # device and a driver ioctl look identical in /proc)
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=f"{len(persistent)} persistent D-state process(es): "
                + ", ".join(f"{b.name}[{b.pid}]" for b in persistent),
                extra_info=extra,
            )
        if fd_pct >= FD_USAGE_PCT_UNHEALTHY or pid_pct >= PID_USAGE_PCT_UNHEALTHY:
            what = (f"file handles {fd_pct:.1f}% of file-max"
                    if fd_pct >= FD_USAGE_PCT_UNHEALTHY
                    else f"PIDs {pid_pct:.1f}% of pid_max")
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"system resource exhaustion imminent: {what}",
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="find and stop the leaking workload",
                    repair_actions=[RepairActionType.CHECK_USER_APP_AND_GPU],
                ),
            )
        if fd_pct >= FD_USAGE_PCT_DEGRADED or pid_pct >= PID_USAGE_PCT_DEGRADED:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=f"system resource pressure: file handles "
                f"{fd_pct:.1f}% / PIDs {pid_pct:.1f}% of their limits",
                extra_info=extra,
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
