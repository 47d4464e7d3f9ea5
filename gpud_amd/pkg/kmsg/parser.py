"""/dev/kmsg record parser.

Reference: pkg/kmsg/watcher.go:292 parseLine — records look like

    priority,sequence,timestamp_us,flags[,...];message
     SUBSYSTEM=...
     DEVICE=...

``priority`` packs syslog facility<<3 | severity. ``timestamp_us`` is
microseconds since boot; wall-clock time is recovered by adding the boot
time (CLOCK_REALTIME − CLOCK_BOOTTIME), same as the reference's boot-time
offset handling.
"""

from __future__ import annotations

import datetime
import time
from dataclasses import dataclass, field
from typing import Dict, Optional


def boot_wall_time() -> float:
    """Wall-clock epoch seconds at which the machine booted."""
    return time.clock_gettime(time.CLOCK_REALTIME) - time.clock_gettime(
        time.CLOCK_BOOTTIME
    )


@dataclass
class Message:
    priority: int = 0
    sequence: int = 0
    timestamp_us: int = 0  # microseconds since boot
    message: str = ""
    extra: Dict[str, str] = field(default_factory=dict)
    # wall-clock timestamp, filled by the parser from the boot offset
    time: Optional[datetime.datetime] = None

    @property
    def facility(self) -> int:
        return self.priority >> 3

    @property
    def severity(self) -> int:
        return self.priority & 7

    def described_severity(self) -> str:
        return _SEVERITIES[self.severity] if 0 <= self.severity < 8 else "unknown"


_SEVERITIES = [
    "emerg", "alert", "crit", "err", "warning", "notice", "info", "debug",
]


def parse_line(line: str, boot_time_epoch: Optional[float] = None) -> Optional[Message]:
    """Parse one /dev/kmsg record line (not a continuation line).

    Returns None for malformed or continuation lines (leading space).
    """
    if not line or line[0] in (" ", "\t"):
        return None
    sep = line.find(";")
    if sep < 0:
        return None
    header, msg = line[:sep], line[sep + 1:].rstrip("\n")
    parts = header.split(",")
    if len(parts) < 3:
        return None
    try:
        priority = int(parts[0])
        sequence = int(parts[1])
        timestamp_us = int(parts[2])
    except ValueError:
        return None
    if boot_time_epoch is None:
        boot_time_epoch = boot_wall_time()
    wall = datetime.datetime.fromtimestamp(
        boot_time_epoch + timestamp_us / 1e6, tz=datetime.timezone.utc
    )
    return Message(
        priority=priority,
        sequence=sequence,
        timestamp_us=timestamp_us,
        message=msg,
        time=wall,
    )


def parse_continuation(line: str, msg: Message) -> bool:
    """Fold a ``' KEY=value'`` continuation line into ``msg.extra``."""
    if not line or line[0] not in (" ", "\t"):
        return False
    body = line.strip()
    if "=" in body:
        k, _, v = body.partition("=")
        msg.extra[k] = v
        return True
    return False
