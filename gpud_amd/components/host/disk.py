"""disk — mount-point usage, mount-target tracking, block devices.

Reference: components/disk + pkg/disk. Mechanism parity items:

  * statfs usage per mount point with used-percent thresholds
    (disk/component.go free-space thresholds);
  * **findmnt with retries** for every tracked mount target
    (disk/component.go:600-623 — findmnt is occasionally flaky and returns
    empty output which fails JSON parsing; that is transient, so up to 5
    attempts with a retry interval, only a fully-exhausted budget is an
    error), ``--target --json --df --bytes``, command override supported
    (the reference's nsenter seam, pkg/disk/findmnt.go:35);
  * **lsblk with flush-retry** and recursive tree flattening
    (disk/component.go:175-181, pkg/disk/lsblk.go) — transient lsblk
    failures retry, children are flattened into the device list, and
    devices whose fstype/mountpoint are missing from lsblk are back-filled
    via findmnt (pkg/disk's fstype fallback).

Every external probe is a function field so tests inject fakes (the
reference's injected-function-field pattern).

Disk-health kernel messages land in THIS component's event bucket
(reference: disk/kmsg_matcher.go — RAID array failure, filesystem
remounted read-only, NVMe path/timeout/disable, beyond-end-of-device,
buffer I/O and superblock write errors).
"""

from __future__ import annotations

import re

import json
import subprocess
import time
from typing import Callable, Dict, List, Optional

import psutil

from ...apiv1.types import HealthStateType
from ...pkg.kmsg.syncer import MatchResult, Syncer
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "disk"

# kmsg matcher (reference: components/disk/kmsg_matcher.go constants)
_KMSG_RULES = (
    ("raid_array_failure",
     re.compile(r"md/raid.*: Disk failure on .* detected, failing array"),
     "Critical"),
    ("filesystem_read_only",
     re.compile(r"Remounting filesystem read-only"), "Critical"),
    ("nvme_path_failure",
     re.compile(r"block nvme.*: no available path - failing I/O"),
     "Critical"),
    ("nvme_controller_timeout",
     re.compile(r"nvme nvme\d+: I/O .* timeout, reset controller"),
     "Warning"),
    ("nvme_device_disabled",
     re.compile(r"nvme nvme\d+: Disabling device after reset failure"),
     "Critical"),
    ("beyond_end_of_device",
     re.compile(r"attempt to access beyond end of device"), "Warning"),
    ("buffer_io_error",
     re.compile(r"Buffer I/O error on dev \S+, logical block \d+"),
     "Warning"),
    ("superblock_write_error",
     re.compile(r"I/O error while writing superblock"), "Critical"),
)


def match_disk_kmsg(line):
    for name, rx, event_type in _KMSG_RULES:
        if rx.search(line):
            return MatchResult(name=name, event_type=event_type, message=line)
    return None

DEFAULT_USED_PERCENT_DEGRADED = 90.0
DEFAULT_USED_PERCENT_UNHEALTHY = 98.0
FINDMNT_RETRIES = 5  # reference: disk/component.go:595 "for attempt := range 5"
LSBLK_RETRIES = 3
DEFAULT_RETRY_INTERVAL_SECONDS = 1.0


def _run(cmd: List[str], timeout: float = 15.0) -> subprocess.CompletedProcess:
    return subprocess.run(cmd, capture_output=True, text=True, timeout=timeout)


# ---------------------------------------------------------------------------
# findmnt (reference: pkg/disk/findmnt.go)
# ---------------------------------------------------------------------------


def find_mnt(
    target: str,
    findmnt_command: str = "",
    run: Callable = _run,
) -> Optional[Dict]:
    """One findmnt invocation for a mount target. Returns the reference's
    FindMntOutput shape ({target, filesystems:[{mounted_point, sources,
    fstype, size_bytes, used_bytes, available_bytes, used_percent}]}) or
    raises on command/parse failure (the caller retries)."""
    cmd = (findmnt_command or "findmnt").split() + [
        "--target", target, "--json", "--df", "--bytes",
    ]
    out = run(cmd)
    if out.returncode != 0:
        raise RuntimeError(f"findmnt exit {out.returncode}: {out.stderr[:200]}")
    raw = json.loads(out.stdout)  # empty/garbled output raises -> retried
    filesystems = []
    for fs in raw.get("filesystems", []):
        sources = fs.get("sources") or []
        if not sources and fs.get("source"):
            sources = [fs["source"]]
        pct_raw = str(fs.get("use%", "0%")).rstrip("%")
        try:
            pct = float(pct_raw)
        except ValueError:
            pct = 0.0
        filesystems.append(
            {
                "mounted_point": fs.get("target", ""),
                "sources": sources,
                "fstype": fs.get("fstype", ""),
                "size_bytes": int(fs.get("size") or 0),
                "used_bytes": int(fs.get("used") or 0),
                "available_bytes": int(fs.get("avail") or 0),
                "used_percent": pct,
            }
        )
    return {"target": target, "filesystems": filesystems}


def find_mnt_with_retries(
    target: str,
    findmnt_command: str = "",
    retries: int = FINDMNT_RETRIES,
    retry_interval: float = DEFAULT_RETRY_INTERVAL_SECONDS,
    run: Callable = _run,
    sleep: Callable = time.sleep,
) -> Optional[Dict]:
    """findmnt with the reference's transient-failure retry loop
    (disk/component.go:600-623). Returns None only after the whole retry
    budget is exhausted."""
    for attempt in range(retries):
        try:
            return find_mnt(target, findmnt_command, run=run)
        except (RuntimeError, OSError, ValueError, subprocess.TimeoutExpired,
                json.JSONDecodeError):
            if attempt + 1 < retries:
                sleep(retry_interval)
    return None


# ---------------------------------------------------------------------------
# lsblk (reference: pkg/disk/lsblk.go — JSON tree, flatten, fstype fallback)
# ---------------------------------------------------------------------------

_LSBLK_COLUMNS = (
    "NAME,TYPE,SIZE,MOUNTPOINT,FSTYPE,PKNAME,ROTA,SERIAL,WWN,VENDOR,MODEL,"
    "REV,FSUSED"
)


def _flatten_devices(devs: List[Dict], parent: str = "") -> List[Dict]:
    """Flatten lsblk's nested children tree (pkg/disk/lsblk_flatten.go):
    every device carries its parent's name so usage can be attributed to
    the physical device."""
    flat: List[Dict] = []
    for d in devs:
        entry = {k: v for k, v in d.items() if k != "children"}
        if parent and not entry.get("pkname"):
            entry["pkname"] = parent
        flat.append(entry)
        if d.get("children"):
            flat.extend(_flatten_devices(d["children"], d.get("name", "")))
    return flat


def list_block_devices(
    lsblk_command: str = "",
    findmnt_command: str = "",
    retries: int = LSBLK_RETRIES,
    retry_interval: float = DEFAULT_RETRY_INTERVAL_SECONDS,
    run: Callable = _run,
    sleep: Callable = time.sleep,
) -> Optional[List[Dict]]:
    """lsblk JSON with the flush-retry loop (transient empty/garbled output
    retries — disk/component.go:175-181) and the fstype fallback: a mounted
    device lsblk reports without an fstype is back-filled via findmnt
    (pkg/disk's DefaultFsTypeFunc fallback)."""
    cmd = (lsblk_command or "lsblk").split() + [
        "-J", "-b", "-o", _LSBLK_COLUMNS,
    ]
    devices: Optional[List[Dict]] = None
    for attempt in range(retries):
        try:
            out = run(cmd)
            if out.returncode != 0:
                raise RuntimeError(f"lsblk exit {out.returncode}")
            devices = json.loads(out.stdout).get("blockdevices", [])
            break
        except (OSError, RuntimeError, subprocess.TimeoutExpired,
                json.JSONDecodeError, ValueError):
            if attempt + 1 < retries:
                sleep(retry_interval)
    if devices is None:
        return None
    flat = _flatten_devices(devices)
    # fstype back-fill for mounted devices lsblk could not type
    for d in flat:
        if d.get("mountpoint") and not d.get("fstype"):
            try:
                mnt = find_mnt(d["mountpoint"], findmnt_command, run=run)
            except Exception:  # noqa: BLE001 — fallback only
                continue
            for fs in mnt["filesystems"]:
                if fs["fstype"]:
                    d["fstype"] = fs["fstype"]
                    break
    return flat


class DiskComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME)
            if inst.event_store is not None else None
        )
        self._kmsg = inst.kmsg_reader
        self._syncer = None
        self.mount_points = list(inst.mount_points or ["/"])
        self.mount_targets = list(inst.mount_targets or [])
        self._lsblk_command = inst.lsblk_command
        self._findmnt_command = getattr(inst, "findmnt_command", "")
        self.retry_interval = DEFAULT_RETRY_INTERVAL_SECONDS
        # injected function fields (reference test pattern)
        self.get_block_devices: Callable = lambda: list_block_devices(
            self._lsblk_command,
            self._findmnt_command,
            retry_interval=self.retry_interval,
        )
        self.find_mnt: Callable = lambda target: find_mnt_with_retries(
            target,
            self._findmnt_command,
            retry_interval=self.retry_interval,
        )
        # cached last probes (queryable via extra_info)
        self.mount_target_usages: Dict[str, Dict] = {}

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def start(self) -> None:
        # disk-health kmsg events into this component's bucket
        # (reference: disk/kmsg_matcher.go wiring in disk/component.go)
        if self._kmsg is not None and self._bucket is not None:
            self._syncer = Syncer(self._kmsg, match_disk_kmsg, self._bucket)
        super().start()

    def events(self, since):
        return self._bucket.get(since) if self._bucket is not None else []

    def check(self) -> CheckResult:
        degraded, unhealthy, missing = [], [], []
        extra = {}
        for mp in self.mount_points:
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

        # mount-target tracking via findmnt-with-retries (reference:
        # MountTargetUsages — a target findmnt cannot resolve after the
        # retry budget is recorded, logged, and does not flip health,
        # matching disk/component.go:626)
        failed_targets = []
        for target in self.mount_targets:
            mnt = self.find_mnt(target)
            if mnt is None:
                failed_targets.append(target)
                continue
            self.mount_target_usages[target] = mnt
            for fs in mnt["filesystems"]:
                extra[f"target.{target}.mounted_point"] = fs["mounted_point"]
                extra[f"target.{target}.fstype"] = fs["fstype"]
                extra[f"target.{target}.used_percent"] = f"{fs['used_percent']:.1f}"
                self._gauges.set(
                    "mount_target_used_percent",
                    "Mount-target used percent (findmnt)",
                    fs["used_percent"],
                    mount_point=target,
                )
                break
        if failed_targets:
            extra["findmnt_failed_targets"] = ", ".join(failed_targets)

        # block-device tree (flush-retried lsblk)
        devs = self.get_block_devices()
        if devs is not None:
            extra["block_devices"] = str(len(devs))
            disks = [d for d in devs if d.get("type") == "disk"]
            extra["disks"] = str(len(disks))

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
