"""docker — docker daemon reachability.

Reference: components/docker (moby client container list —
pkg/docker/docker.go:7-8). Python-first: docker unix-socket HTTP ping
(GET /_ping) without a client library dependency.
"""

from __future__ import annotations

import http.client
import os
import socket
from typing import Callable, Optional

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "docker"

DEFAULT_SOCKET = "/var/run/docker.sock"


class _UnixHTTPConnection(http.client.HTTPConnection):
    def __init__(self, path: str, timeout: float = 3.0):
        super().__init__("localhost", timeout=timeout)
        self._path = path

    def connect(self) -> None:
        s = socket.socket(socket.AF_UNIX, socket.SOCK_STREAM)
        s.settimeout(self.timeout)
        s.connect(self._path)
        self.sock = s


def docker_ping(socket_path: str = DEFAULT_SOCKET) -> Optional[str]:
    """Returns the API version header on success, None on failure."""
    if not os.path.exists(socket_path):
        return None
    try:
        conn = _UnixHTTPConnection(socket_path)
        conn.request("GET", "/_ping")
        resp = conn.getresponse()
        if resp.status == 200:
            return resp.getheader("Api-Version", "unknown")
    except OSError:
        pass
    return None


class DockerComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        cfg = inst.config
        self.socket_path = (
            getattr(cfg, "docker_socket", DEFAULT_SOCKET) if cfg else DEFAULT_SOCKET
        )
        self.ping: Callable = lambda: docker_ping(self.socket_path)

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME, "container"]

    def is_supported(self) -> bool:
        return os.path.exists(self.socket_path)

    def check(self) -> CheckResult:
        ver = self.ping()
        if ver is None:
            if not os.path.exists(self.socket_path):
                return CheckResult(
                    NAME, reason=f"docker socket absent ({self.socket_path})"
                )
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"docker socket present but not responding ({self.socket_path})",
            )
        return CheckResult(
            NAME,
            reason=f"docker daemon responding (API {ver})",
            extra_info={"api_version": ver},
        )


def new(inst: GPUdInstance) -> Component:
    return DockerComponent(inst)
