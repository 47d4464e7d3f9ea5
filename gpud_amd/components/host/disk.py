"""disk — mount-point usage and block devices.

Reference: components/disk (statfs usage per mount point, findmnt/lsblk
with nsenter overrides and retries — disk/component.go:600-623,175-181).
Python-first: psutil statfs for usage; ``lsblk -J`` (override-able command)
for the block-device tree.
"""

from __future__ import annotations

import json
import subprocess
from typing import Callable, Dict, List, Optional

import psutil

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "disk"

DEFAULT_USED_PERCENT_DEGRADED = 90.0
DEFAULT_USED_PERCENT_UNHEALTHY = 98.0


def list_block_devices(lsblk_command: str = "") -> Optional[List[Dict]]:
    cmd = lsblk_command or "lsblk"
    try:
        out = subprocess.run(
            [cmd, "-J", "-b", "-o", "NAME,TYPE,SIZE,MOUNTPOINT,FSTYPE"],
            capture_output=True,
            text=True,
            timeout=15,
        )
        if out.returncode != 0:
            return None
        return json.loads(out.stdout).get("blockdevices", [])
    except (OSError, subprocess.TimeoutExpired, json.JSONDecodeError):
        return None


class DiskComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self.mount_points = list(inst.mount_points or ["/"])
        self.mount_targets = list(inst.mount_targets or [])
        self._lsblk_command = inst.lsblk_command
        self.get_block_devices: Callable = lambda: list_block_devices(
            self._lsblk_command
        )

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def check(self) -> CheckResult:
        degraded, unhealthy, missing = [], [], []
        extra = {}
        for mp in self.mount_points + self.mount_targets:
            try:
                u = psutil.disk_usage(mp)
            except OSError:
                missing.append(mp)
                continue
            self._gauges.set(
                "disk_total_bytes", "Filesystem size", u.total, mount_point=mp
            )
            self._gauges.set(
                "disk_used_bytes", "Filesystem used bytes", u.used, mount_point=mp
            )
            self._gauges.set(
                "disk_used_percent", "Filesystem used percent", u.percent,
                mount_point=mp,
            )
            extra[f"{mp}.used_percent"] = f"{u.percent:.1f}"
            if u.percent >= DEFAULT_USED_PERCENT_UNHEALTHY:
                unhealthy.append(mp)
            elif u.percent >= DEFAULT_USED_PERCENT_DEGRADED:
                degraded.append(mp)
        if unhealthy or missing:
            parts = []
            if unhealthy:
                parts.append("nearly full: " + ", ".join(unhealthy))
            if missing:
                parts.append("unreadable mount points: " + ", ".join(missing))
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="; ".join(parts),
                extra_info=extra,
            )
        if degraded:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason="filesystems over "
                f"{DEFAULT_USED_PERCENT_DEGRADED:.0f}%: " + ", ".join(degraded),
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"{len(self.mount_points)} mount point(s) healthy",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return DiskComponent(inst)
