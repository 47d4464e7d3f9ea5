"""Wire types for the gpud_amd v1 API.

Field names and enum values are kept wire-compatible with the reference's
``api/v1/types.go`` (reference: api/v1/types.go:50 HealthState,
:108 Event, :136 Metric, :183 RepairActionType, :222 EventType, :261
MachineInfo) so existing gpud clients/control planes can consume the
responses unchanged. Implemented as plain dataclasses with explicit
``to_dict``/``from_dict`` — these types sit on the serving path of every
``/v1/*`` request, so we avoid per-request pydantic model validation.
"""

from __future__ import annotations

import datetime
from dataclasses import dataclass, field
from typing import Any, Dict, List, Optional


# ---------------------------------------------------------------------------
# Enums (string-valued, matching reference api/v1/types.go)
# ---------------------------------------------------------------------------

class HealthStateType:
    HEALTHY = "Healthy"
    UNHEALTHY = "Unhealthy"
    DEGRADED = "Degraded"
    INITIALIZING = "Initializing"


class RunModeType:
    AUTO = "auto"
    MANUAL = "manual"


class ComponentType:
    CUSTOM_PLUGIN = "custom-plugin"


class EventType:
    UNKNOWN = "Unknown"
    INFO = "Info"
    WARNING = "Warning"
    CRITICAL = "Critical"
    FATAL = "Fatal"

    _ALL = ("Info", "Warning", "Critical", "Fatal")

    @staticmethod
    def from_string(s: str) -> str:
        return s if s in EventType._ALL else EventType.UNKNOWN


class RepairActionType:
    IGNORE_NO_ACTION_REQUIRED = "IGNORE_NO_ACTION_REQUIRED"
    REBOOT_SYSTEM = "REBOOT_SYSTEM"
    HARDWARE_INSPECTION = "HARDWARE_INSPECTION"
    CHECK_USER_APP_AND_GPU = "CHECK_USER_APP_AND_GPU"


class PackagePhase:
    INSTALLED = "Installed"
    INSTALLING = "Installing"
    UNKNOWN = "Unknown"
    SKIPPED = "Skipped"


# ---------------------------------------------------------------------------
# Time helpers — RFC3339 like k8s metav1.Time
# ---------------------------------------------------------------------------

def rfc3339(t: Optional[datetime.datetime]) -> Optional[str]:
    if t is None:
        return None
    if t.tzinfo is None:
        t = t.replace(tzinfo=datetime.timezone.utc)
    # metav1.Time marshals at second precision
    return t.astimezone(datetime.timezone.utc).strftime("%Y-%m-%dT%H:%M:%SZ")


def parse_rfc3339(s: Optional[str]) -> Optional[datetime.datetime]:
    if not s:
        return None
    s = s.replace("Z", "+00:00")
    return datetime.datetime.fromisoformat(s)


def utcnow() -> datetime.datetime:
    return datetime.datetime.now(datetime.timezone.utc)


# ---------------------------------------------------------------------------
# Core types
# ---------------------------------------------------------------------------

@dataclass
class SuggestedActions:
    """Reference: api/v1/types.go:207 SuggestedActions."""

    description: str = ""
    repair_actions: List[str] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "description": self.description,
            "repair_actions": list(self.repair_actions),
        }

    @staticmethod
    def from_dict(d: Optional[Dict[str, Any]]) -> Optional["SuggestedActions"]:
        if d is None:
            return None
        return SuggestedActions(
            description=d.get("description", ""),
            repair_actions=list(d.get("repair_actions") or []),
        )

    def describe_actions(self) -> str:
        return ", ".join(self.repair_actions)


@dataclass
class HealthState:
    """Reference: api/v1/types.go:50 HealthState."""

    time: datetime.datetime = field(default_factory=utcnow)
    component: str = ""
    component_type: str = ""
    name: str = ""
    run_mode: str = ""
    health: str = HealthStateType.HEALTHY
    reason: str = ""
    error: str = ""
    suggested_actions: Optional[SuggestedActions] = None
    extra_info: Optional[Dict[str, str]] = None
    raw_output: str = ""

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"time": rfc3339(self.time)}
        if self.component:
            d["component"] = self.component
        if self.component_type:
            d["component_type"] = self.component_type
        if self.name:
            d["name"] = self.name
        if self.run_mode:
            d["run_mode"] = self.run_mode
        if self.health:
            d["health"] = self.health
        if self.reason:
            d["reason"] = self.reason
        if self.error:
            d["error"] = self.error
        if self.suggested_actions is not None:
            d["suggested_actions"] = self.suggested_actions.to_dict()
        if self.extra_info:
            d["extra_info"] = dict(self.extra_info)
        if self.raw_output:
            d["raw_output"] = self.raw_output[:4096]
        return d

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "HealthState":
        return HealthState(
            time=parse_rfc3339(d.get("time")) or utcnow(),
            component=d.get("component", ""),
            component_type=d.get("component_type", ""),
            name=d.get("name", ""),
            run_mode=d.get("run_mode", ""),
            health=d.get("health", ""),
            reason=d.get("reason", ""),
            error=d.get("error", ""),
            suggested_actions=SuggestedActions.from_dict(d.get("suggested_actions")),
            extra_info=d.get("extra_info"),
            raw_output=d.get("raw_output", ""),
        )


@dataclass
class Event:
    """Reference: api/v1/types.go:108 Event."""

    time: datetime.datetime = field(default_factory=utcnow)
    component: str = ""
    name: str = ""
    type: str = EventType.INFO
    message: str = ""

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {"time": rfc3339(self.time)}
        if self.component:
            d["component"] = self.component
        if self.name:
            d["name"] = self.name
        if self.type:
            d["type"] = self.type
        if self.message:
            d["message"] = self.message
        return d

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Event":
        return Event(
            time=parse_rfc3339(d.get("time")) or utcnow(),
            component=d.get("component", ""),
            name=d.get("name", ""),
            type=d.get("type", EventType.UNKNOWN),
            message=d.get("message", ""),
        )


@dataclass
class Metric:
    """Reference: api/v1/types.go:136 Metric."""

    unix_seconds: int = 0
    name: str = ""
    labels: Optional[Dict[str, str]] = None
    value: float = 0.0

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {
            "unix_seconds": self.unix_seconds,
            "name": self.name,
            "value": self.value,
        }
        if self.labels:
            d["labels"] = dict(self.labels)
        return d

    @staticmethod
    def from_dict(d: Dict[str, Any]) -> "Metric":
        return Metric(
            unix_seconds=int(d.get("unix_seconds", 0)),
            name=d.get("name", ""),
            labels=d.get("labels"),
            value=float(d.get("value", 0.0)),
        )


# ---------------------------------------------------------------------------
# Aggregate response types (GPUdComponent* in the reference)
# ---------------------------------------------------------------------------

@dataclass
class ComponentHealthStates:
    component: str = ""
    states: List[HealthState] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "component": self.component,
            "states": [s.to_dict() for s in self.states],
        }


@dataclass
class ComponentEvents:
    component: str = ""
    start_time: Optional[datetime.datetime] = None
    end_time: Optional[datetime.datetime] = None
    events: List[Event] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "component": self.component,
            "startTime": rfc3339(self.start_time),
            "endTime": rfc3339(self.end_time),
            "events": [e.to_dict() for e in self.events],
        }


@dataclass
class ComponentMetrics:
    component: str = ""
    metrics: List[Metric] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "component": self.component,
            "metrics": [m.to_dict() for m in self.metrics],
        }


@dataclass
class Info:
    states: List[HealthState] = field(default_factory=list)
    events: List[Event] = field(default_factory=list)
    metrics: List[Metric] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "states": [s.to_dict() for s in self.states],
            "events": [e.to_dict() for e in self.events],
            "metrics": [m.to_dict() for m in self.metrics],
        }


@dataclass
class ComponentInfo:
    component: str = ""
    start_time: Optional[datetime.datetime] = None
    end_time: Optional[datetime.datetime] = None
    info: Info = field(default_factory=Info)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "component": self.component,
            "startTime": rfc3339(self.start_time),
            "endTime": rfc3339(self.end_time),
            "info": self.info.to_dict(),
        }


@dataclass
class PackageStatus:
    name: str = ""
    phase: str = PackagePhase.UNKNOWN
    status: str = ""
    current_version: str = ""

    def to_dict(self) -> Dict[str, Any]:
        return {
            "name": self.name,
            "phase": self.phase,
            "status": self.status,
            "current_version": self.current_version,
        }


# ---------------------------------------------------------------------------
# Machine info (reference: api/v1/types.go:261 MachineInfo) — AMD-first: the
# GPU driver is amdgpu, the compute stack version is ROCm/HIP (the reference's
# cudaVersion slot carries the ROCm version string for control-plane compat).
# ---------------------------------------------------------------------------

@dataclass
class MachineGPUInstance:
    uuid: str = ""
    sn: str = ""
    min_power: int = 0
    max_power: int = 0
    product: str = ""
    board_id: int = 0

    def to_dict(self) -> Dict[str, Any]:
        return {
            "uuid": self.uuid,
            "sn": self.sn,
            "minPower": self.min_power,
            "maxPower": self.max_power,
            "product": self.product,
            "boardID": self.board_id,
        }


@dataclass
class MachineGPUInfo:
    product: str = ""
    manufacturer: str = ""
    architecture: str = ""
    driver_version: str = ""
    rocm_version: str = ""
    memory: str = ""
    gpus: List[MachineGPUInstance] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {
            "product": self.product,
            "manufacturer": self.manufacturer,
            "architecture": self.architecture,
            "driverVersion": self.driver_version,
            "rocmVersion": self.rocm_version,
            "memory": self.memory,
            "gpus": [g.to_dict() for g in self.gpus],
        }


@dataclass
class MachineCPUInfo:
    type: str = ""
    manufacturer: str = ""
    architecture: str = ""
    logical_cores: int = 0

    def to_dict(self) -> Dict[str, Any]:
        return {
            "type": self.type,
            "manufacturer": self.manufacturer,
            "architecture": self.architecture,
            "logicalCores": self.logical_cores,
        }


@dataclass
class MachineMemoryInfo:
    total_bytes: int = 0

    def to_dict(self) -> Dict[str, Any]:
        return {"totalBytes": self.total_bytes}


@dataclass
class MachineDiskInfo:
    block_devices: List[Dict[str, Any]] = field(default_factory=list)
    container_root_disk: str = ""

    def to_dict(self) -> Dict[str, Any]:
        return {
            "blockDevices": self.block_devices,
            "containerRootDisk": self.container_root_disk,
        }


@dataclass
class MachineNICInfo:
    private_ip_interfaces: List[Dict[str, Any]] = field(default_factory=list)

    def to_dict(self) -> Dict[str, Any]:
        return {"privateIPInterfaces": self.private_ip_interfaces}


@dataclass
class MachineInfo:
    gpud_version: str = ""
    gpu_driver_version: str = ""
    cuda_version: str = ""  # carries ROCm version on AMD (wire-compat slot)
    container_runtime_version: str = ""
    tailscale_version: str = ""  # reference: MachineInfo.TailscaleVersion
    kernel_version: str = ""
    os_image: str = ""
    operating_system: str = ""
    system_uuid: str = ""
    machine_id: str = ""
    boot_id: str = ""
    hostname: str = ""
    uptime: Optional[datetime.datetime] = None
    cpu_info: Optional[MachineCPUInfo] = None
    memory_info: Optional[MachineMemoryInfo] = None
    gpu_info: Optional[MachineGPUInfo] = None
    disk_info: Optional[MachineDiskInfo] = None
    nic_info: Optional[MachineNICInfo] = None

    def to_dict(self) -> Dict[str, Any]:
        d: Dict[str, Any] = {}
        if self.gpud_version:
            d["gpudVersion"] = self.gpud_version
        if self.gpu_driver_version:
            d["gpuDriverVersion"] = self.gpu_driver_version
        if self.cuda_version:
            d["cudaVersion"] = self.cuda_version
        if self.container_runtime_version:
            d["containerRuntimeVersion"] = self.container_runtime_version
        if self.tailscale_version:
            d["tailscaleVersion"] = self.tailscale_version
        if self.kernel_version:
            d["kernelVersion"] = self.kernel_version
        if self.os_image:
            d["osImage"] = self.os_image
        if self.operating_system:
            d["operatingSystem"] = self.operating_system
        if self.system_uuid:
            d["systemUUID"] = self.system_uuid
        if self.machine_id:
            d["machineID"] = self.machine_id
        if self.boot_id:
            d["bootID"] = self.boot_id
        if self.hostname:
            d["hostname"] = self.hostname
        if self.uptime is not None:
            d["uptime"] = rfc3339(self.uptime)
        if self.cpu_info is not None:
            d["cpuInfo"] = self.cpu_info.to_dict()
        if self.memory_info is not None:
            d["memoryInfo"] = self.memory_info.to_dict()
        if self.gpu_info is not None:
            d["gpuInfo"] = self.gpu_info.to_dict()
        if self.disk_info is not None:
            d["diskInfo"] = self.disk_info.to_dict()
        if self.nic_info is not None:
            d["nicInfo"] = self.nic_info.to_dict()
        return d
