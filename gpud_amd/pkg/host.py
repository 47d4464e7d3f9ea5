"""Host identity and the reboot event store.

Reference: pkg/host/event.go:22-142 RebootEventStore — records a ``reboot``
event at each boot boundary (detected from boot time = now − uptime); error
components (RAS, throttle) consume it to reset their health state machines
after a reboot, and the suggested-action escalation counts reboots
(reference: xid/health_state.go:61-97).
"""

from __future__ import annotations

import datetime
import os
import platform
import socket
import time
from typing import List, Optional

from ..apiv1.types import Event, EventType
from .eventstore import Store
from .log import logger

REBOOT_BUCKET = "os"
EVENT_NAME_REBOOT = "reboot"  # reference: pkg/host/event.go:17
# a new boot is recorded only if it differs from the last recorded boot by
# more than this margin (uptime reads are not exact)
BOOT_TIME_MARGIN_SECONDS = 30


def boot_time() -> datetime.datetime:
    bt = time.clock_gettime(time.CLOCK_REALTIME) - time.clock_gettime(
        time.CLOCK_BOOTTIME
    )
    return datetime.datetime.fromtimestamp(bt, tz=datetime.timezone.utc)


def uptime_seconds() -> float:
    return time.clock_gettime(time.CLOCK_BOOTTIME)


def boot_id() -> str:
    try:
        with open("/proc/sys/kernel/random/boot_id") as f:
            return f.read().strip()
    except OSError:
        return ""


def machine_id() -> str:
    for p in ("/etc/machine-id", "/var/lib/dbus/machine-id"):
        try:
            with open(p) as f:
                return f.read().strip()
        except OSError:
            continue
    return ""


def system_uuid() -> str:
    try:
        with open("/sys/class/dmi/id/product_uuid") as f:
            return f.read().strip()
    except OSError:
        return ""


def hostname() -> str:
    return socket.gethostname()


def kernel_version() -> str:
    return platform.release()


def os_image() -> str:
    try:
        with open("/etc/os-release") as f:
            for line in f:
                if line.startswith("PRETTY_NAME="):
                    return line.split("=", 1)[1].strip().strip('"')
    except OSError:
        pass
    return ""


class RebootEventStore:
    def __init__(self, event_store: Store):
        self._bucket = event_store.bucket(REBOOT_BUCKET)

    def record_reboot(self) -> Optional[Event]:
        """Record the current boot as a reboot event if not yet recorded."""
        bt = boot_time()
        recent = self._bucket.find_by_name_since(
            EVENT_NAME_REBOOT,
            bt - datetime.timedelta(seconds=BOOT_TIME_MARGIN_SECONDS),
        )
        for ev in recent:
            if abs((ev.time - bt).total_seconds()) <= BOOT_TIME_MARGIN_SECONDS:
                return None  # this boot already recorded
        ev = Event(
            time=bt,
            component=REBOOT_BUCKET,
            name=EVENT_NAME_REBOOT,
            type=EventType.WARNING,
            message=f"system boot detected (boot id {boot_id() or 'unknown'})",
        )
        self._bucket.insert(ev)
        logger.info("recorded reboot event at %s", bt.isoformat())
        return ev

    def get_reboot_events(self, since: datetime.datetime) -> List[Event]:
        return self._bucket.find_by_name_since(EVENT_NAME_REBOOT, since)

    def reboot_count_since(self, since: datetime.datetime) -> int:
        return len(self.get_reboot_events(since))


def reboot_machine(reboot_command: str = "") -> Optional[str]:
    """Run the (configurable) reboot command (reference: pkg/host/reboot.go:47)."""
    cmd = reboot_command or "reboot"
    rc = os.system(cmd)
    if rc != 0:
        return f"reboot command {cmd!r} exited {rc}"
    return None
