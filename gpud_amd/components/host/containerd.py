"""containerd — container runtime health.

Reference: components/containerd (socket existence with consecutive-miss
threshold, service active, CRI version probe — containerd/component.go:28).
Python-first: unix-socket existence + connect probe + systemctl state;
the consecutive-miss threshold avoids flapping on restarts
(reference: components/registry.go:105-109).
"""

from __future__ import annotations

import os
import socket
import subprocess
from typing import Callable

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "containerd"

DEFAULT_SOCKET = "/run/containerd/containerd.sock"
MISS_THRESHOLD = 3  # consecutive misses before unhealthy


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
            return CheckResult(
                NAME,
                reason=f"containerd socket connectable ({self.socket_path})",
                extra_info={"service": svc},
            )
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
