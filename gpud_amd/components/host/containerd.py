"""containerd — container runtime health.

Reference: components/containerd (socket existence with consecutive-miss
threshold, service active, CRI gRPC version probe, GPU runtime-config
presence in /etc/containerd/config.toml — containerd/component.go:28,283).
Python-first: unix-socket existence + connect probe + systemctl state; a
raw-bytes CRI ``runtime.v1.RuntimeService/Version`` gRPC call (hand-framed
protobuf, no generated stubs — the same approach as session/protowire);
the consecutive-miss threshold avoids flapping on restarts
(reference: components/registry.go:105-109).
"""

from __future__ import annotations

import os
import socket
import subprocess
from typing import Callable, Dict, Optional

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "containerd"

DEFAULT_SOCKET = "/run/containerd/containerd.sock"
DEFAULT_CONFIG_PATH = "/etc/containerd/config.toml"
MISS_THRESHOLD = 3  # consecutive misses before unhealthy

# GPU runtime plugin paths in containerd's config.toml. containerd 1.x uses
# the io.containerd.grpc.v1.cri plugin path; 2.x moved to
# io.containerd.cri.v1.runtime (reference: containerd/component.go:36-40).
# The AMD container stack registers an "amd" runtime the same way the
# nvidia-container-toolkit registers "nvidia".
GPU_RUNTIME_MARKERS = (
    'plugins."io.containerd.grpc.v1.cri".containerd.runtimes.amd',
    'plugins."io.containerd.cri.v1.runtime".containerd.runtimes.amd',
    'plugins."io.containerd.grpc.v1.cri".containerd.runtimes.nvidia',
    'plugins."io.containerd.cri.v1.runtime".containerd.runtimes.nvidia',
)


def _decode_string_fields(data: bytes) -> Dict[int, str]:
    """Minimal protobuf decode: field_no -> utf-8 string for LEN fields."""
    out: Dict[int, str] = {}
    pos = 0
    while pos < len(data):
        tag = 0
        shift = 0
        while True:
            b = data[pos]
            pos += 1
            tag |= (b & 0x7F) << shift
            if not b & 0x80:
                break
            shift += 7
        field_no, wire = tag >> 3, tag & 7
        if wire == 2:  # LEN
            ln = 0
            shift = 0
            while True:
                b = data[pos]
                pos += 1
                ln |= (b & 0x7F) << shift
                if not b & 0x80:
                    break
                shift += 7
            out[field_no] = data[pos : pos + ln].decode("utf-8", "replace")
            pos += ln
        elif wire == 0:  # VARINT
            while data[pos] & 0x80:
                pos += 1
            pos += 1
        elif wire == 1:
            pos += 8
        elif wire == 5:
            pos += 4
        else:
            break
    return out


def cri_version(socket_path: str, timeout: float = 10.0) -> Optional[Dict[str, str]]:
    """CRI ``runtime.v1.RuntimeService/Version`` over the containerd socket
    (reference: containerd/component.go:283 CheckVersion). Returns
    {version, runtime_name, runtime_version, runtime_api_version} or None
    when CRI is unreachable/not enabled."""
    try:
        import grpc

        channel = grpc.insecure_channel(f"unix://{socket_path}")
        try:
            call = channel.unary_unary(
                "/runtime.v1.RuntimeService/Version",
                request_serializer=lambda b: b,
                response_deserializer=lambda b: b,
            )
            resp = call(b"", timeout=timeout)  # empty VersionRequest
        finally:
            channel.close()
    except Exception:
        return None
    fields = _decode_string_fields(resp)
    # VersionResponse: 1=version 2=runtime_name 3=runtime_version
    # 4=runtime_api_version (k8s cri-api runtime/v1/api.proto)
    return {
        "version": fields.get(1, ""),
        "runtime_name": fields.get(2, ""),
        "runtime_version": fields.get(3, ""),
        "runtime_api_version": fields.get(4, ""),
    }


def has_gpu_runtime_configuration(config_text: str) -> bool:
    """True when containerd's config declares a GPU runtime under either
    the 1.x or 2.x CRI plugin path (reference:
    hasNvidiaRuntimeConfiguration, containerd/component.go)."""
    compact = config_text.replace(" ", "").replace("'", '"')
    return any(m.replace(" ", "") in compact for m in GPU_RUNTIME_MARKERS)


def read_containerd_config(path: str = DEFAULT_CONFIG_PATH) -> str:
    try:
        with open(path) as f:
            return f.read()
    except OSError:
        return ""


def socket_connectable(path: str, timeout: float = 2.0) -> bool:
    if not os.path.exists(path):
        return False
    s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
    s.settimeout(timeout)
    try:
        s.connect(path)
        return True
    except OSError:
        return False
    finally:
        s.close()


def service_active(name: str = "containerd") -> str:
    try:
        out = subprocess.run(
            ["systemctl", "is-active", name],
            capture_output=True,
            text=True,
            timeout=10,
        )
        return out.stdout.strip()
    except (OSError, subprocess.TimeoutExpired):
        return "unknown"


class ContainerdComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self.socket_path = inst.containerd_address or DEFAULT_SOCKET
        self._misses = 0
        self.check_socket: Callable = lambda: socket_connectable(self.socket_path)
        self.check_service: Callable = service_active
        self.get_cri_version: Callable = lambda: cri_version(self.socket_path)
        self.get_config: Callable = read_containerd_config

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME, "container"]

    def is_supported(self) -> bool:
        # meaningful only on hosts that have (had) containerd installed
        return os.path.exists(self.socket_path) or os.path.exists(
            "/etc/containerd"
        )

    def check(self) -> CheckResult:
        ok = self.check_socket()
        if ok:
            self._misses = 0
            svc = self.check_service()
            extra = {"service": svc}
            # CRI gRPC version probe (reference: containerd active ->
            # CRI version; an unreachable CRI on an active containerd is
            # "installed and active but CRI not enabled" and stays healthy)
            cri = self.get_cri_version()
            if cri is not None and cri.get("runtime_name"):
                extra.update({f"cri_{k}": v for k, v in cri.items() if v})
                reason = (
                    f"containerd active, CRI {cri['runtime_name']} "
                    f"{cri['runtime_version']}"
                )
            else:
                reason = (
                    "containerd socket connectable but CRI is not enabled "
                    f"({self.socket_path})"
                )
            # GPU runtime-config presence (reference: config.toml checked
            # for the nvidia runtime under the 1.x/2.x CRI plugin paths;
            # informational here — AMD GPUs reach containers through
            # /dev/kfd device mounts even without a dedicated runtime)
            config_text = self.get_config()
            if config_text:
                extra["gpu_runtime_configured"] = str(
                    has_gpu_runtime_configuration(config_text)
                ).lower()
            return CheckResult(NAME, reason=reason, extra_info=extra)
        self._misses += 1
        if self._misses >= MISS_THRESHOLD:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"containerd socket missing/unconnectable for "
                f"{self._misses} consecutive checks ({self.socket_path})",
            )
        return CheckResult(
            NAME,
            health=HealthStateType.DEGRADED,
            reason=f"containerd socket miss {self._misses}/{MISS_THRESHOLD}",
        )


def new(inst: GPUdInstance) -> Component:
    return ContainerdComponent(inst)
