"""Daemon configuration (reference: pkg/config/config.go:18-98).

A dataclass covering the reference Config surface: listen address, data dir,
state file, retention periods, compact period, pprof, auto-update, command
overrides, NFS group configs, plugin specs file, component enable/disable
lists — plus AMD-specific knobs (expected GPU count, RAS thresholds).
"""

from __future__ import annotations

import dataclasses
import os
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional

import yaml

DEFAULT_ADDRESS = "localhost:15132"  # reference: cmd/gpud/main.go:15
DEFAULT_DATA_DIR = "/var/lib/gpud"  # reference: pkg/config/default.go:18
DEFAULT_RETENTION_DAYS = 3
DEFAULT_COMPACT_PERIOD_HOURS = 24


@dataclass
class Config:
    address: str = DEFAULT_ADDRESS
    data_dir: str = DEFAULT_DATA_DIR
    state_file: str = ""  # default: <data_dir>/gpud.state
    events_retention_days: float = DEFAULT_RETENTION_DAYS
    metrics_retention_days: float = DEFAULT_RETENTION_DAYS
    compact_period_hours: float = DEFAULT_COMPACT_PERIOD_HOURS
    pprof: bool = False
    enable_auto_update: bool = False
    auto_update_exit_code: int = -1
    poll_interval_seconds: float = 60.0
    metrics_sync_interval_seconds: float = 60.0
    # component selection (reference: component enable/disable list)
    enabled_components: List[str] = field(default_factory=list)
    disabled_components: List[str] = field(default_factory=list)
    # nsenter-style command overrides (test seams)
    reboot_command: str = ""
    findmnt_command: str = ""
    lsblk_command: str = ""
    df_command: str = ""
    lspci_command: str = ""
    containerd_address: str = "/run/containerd/containerd.sock"
    docker_socket: str = "/var/run/docker.sock"
    nfs_host_root: str = ""
    plugin_specs_file: str = ""
    # kmsg source/sink path. Pointing this at a regular file turns the
    # fault-injection loop into a file seam (FileSeamWriter + poll-follow
    # watcher) for environments whose /dev/kmsg write path is rate-limited
    # — same role as the reference's nsenter-style command override seams.
    kmsg_path: str = "/dev/kmsg"
    kernel_modules_to_check: List[str] = field(default_factory=list)
    libraries_to_check: Dict[str, List[str]] = field(default_factory=dict)
    mount_points: List[str] = field(default_factory=lambda: ["/"])
    # accelerator knobs
    expected_gpu_count: int = 0
    expected_xgmi_link_count: int = 0  # 7 per GPU on an 8-GPU MI355X node
    # partition-mode policy: empty = informational only; set to e.g.
    # "SPX"/"NPS1" to make a mismatch Unhealthy (accelerator-amd-partition)
    expected_compute_partition: str = ""
    expected_memory_partition: str = ""
    expected_ib_ports: int = 0
    expected_ib_rate_gbps: float = 0.0
    # network-latency probe targets: list of (host, port)
    latency_targets: List[Any] = field(default_factory=list)
    temperature_margin_threshold_c: float = 10.0
    # error-ras escalation: reboots tolerated before HARDWARE_INSPECTION
    # (reference: xid-reboot-threshold flag, cmd/gpud/run/command.go)
    ras_reboot_threshold: int = 2
    # per-event-name overrides (reference: xid-thresholds per-code map),
    # e.g. {"amdgpu_ring_timeout": 1} escalates that class after 1 reboot
    ras_event_thresholds: Dict[str, int] = field(default_factory=dict)
    zombie_degraded_threshold: int = 1000
    zombie_unhealthy_threshold: int = 2000
    # D-state (uninterruptible sleep) persistence tracking (reference:
    # components/os/threshold.go): consecutive one-minute checks before a
    # blocked process counts as persistent, and the name regexes that gate
    # escalation to unhealthy+reboot (empty list = no escalation)
    dstate_persistence_threshold: int = 5
    dstate_name_regexes: List[str] = field(
        default_factory=lambda: ["^amd", "^rocm"]
    )
    # control plane
    endpoint: str = ""
    token: str = ""
    machine_id: str = ""
    session_protocol: str = "auto"  # v1 | v2 | auto (reference protocol.go)
    # TLS verification towards the control plane (reference:
    # session_v2.go:278 sets MinVersion/ServerName and never skips
    # verification). Verified by default; an operator can pin a private CA
    # bundle, or explicitly opt into insecure mode for lab setups.
    control_plane_insecure_tls: bool = False
    control_plane_ca_file: str = ""

    def control_plane_verify(self):
        """httpx-style ``verify`` value for control-plane connections:
        a CA bundle path when pinned, False only when explicitly opted
        into insecure mode, True (system CAs) otherwise."""
        if self.control_plane_insecure_tls:
            return False
        if self.control_plane_ca_file:
            return self.control_plane_ca_file
        return True

    @property
    def state_path(self) -> str:
        return self.state_file or os.path.join(self.data_dir, "gpud.state")

    @property
    def fifo_path(self) -> str:
        return os.path.join(self.data_dir, "gpud.fifo")

    @property
    def packages_dir(self) -> str:
        return os.path.join(self.data_dir, "packages")

    @property
    def target_version_path(self) -> str:
        return os.path.join(self.data_dir, "target_version")

    def component_enabled(self, name: str, tags: Optional[List[str]] = None) -> bool:
        if self.disabled_components and name in self.disabled_components:
            return False
        if self.enabled_components:
            return name in self.enabled_components
        return True

    def to_dict(self) -> Dict[str, Any]:
        return dataclasses.asdict(self)

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Config":
        known = {f.name for f in dataclasses.fields(Config)}
        return Config(**{k: v for k, v in d.items() if k in known})

    @staticmethod
    def load(path: str) -> "Config":
        with open(path) as f:
            return Config.from_dict(yaml.safe_load(f) or {})

    def save(self, path: str) -> None:
        os.makedirs(os.path.dirname(os.path.abspath(path)), exist_ok=True)
        with open(path, "w") as f:
            yaml.safe_dump(self.to_dict(), f, sort_keys=True)


def default_config(data_dir: str = DEFAULT_DATA_DIR) -> Config:
    return Config(data_dir=data_dir)
