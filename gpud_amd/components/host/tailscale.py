"""tailscale — tailscaled presence, service state, backend state.

Reference: components/tailscale (tailscale/component.go:114-145 —
not-installed is Healthy; installed + service inactive is Unhealthy;
service active but `tailscale status` BackendState != "Running" is
Unhealthy).
"""

from __future__ import annotations

import json
import shutil
import subprocess
from typing import Callable, Optional

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "tailscale"


def tailscale_version() -> Optional[str]:
    path = shutil.which("tailscale")
    if path is None:
        return None
    try:
        out = subprocess.run(
            [path, "version"], capture_output=True, text=True, timeout=10
        )
        return out.stdout.splitlines()[0].strip() if out.stdout else ""
    except (OSError, subprocess.TimeoutExpired, IndexError):
        return ""


def tailscaled_service_active() -> Optional[bool]:
    """systemctl is-active tailscaled; None when systemctl is absent."""
    path = shutil.which("systemctl")
    if path is None:
        return None
    try:
        out = subprocess.run(
            [path, "is-active", "tailscaled"],
            capture_output=True, text=True, timeout=10,
        )
        return out.stdout.strip() == "active"
    except (OSError, subprocess.TimeoutExpired):
        return None


def tailscale_backend_state() -> Optional[str]:
    """BackendState from `tailscale status --json` ("Running",
    "Stopped", "NeedsLogin", ...); None on error."""
    path = shutil.which("tailscale")
    if path is None:
        return None
    try:
        out = subprocess.run(
            [path, "status", "--json"],
            capture_output=True, text=True, timeout=15,
        )
        if out.returncode != 0 or not out.stdout.strip():
            return None
        return json.loads(out.stdout).get("BackendState")
    except (OSError, subprocess.TimeoutExpired, json.JSONDecodeError):
        return None


class TailscaleComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self.get_version: Callable = tailscale_version
        self.get_service_active: Callable = tailscaled_service_active
        self.get_backend_state: Callable = tailscale_backend_state

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME, "network"]

    def is_supported(self) -> bool:
        return shutil.which("tailscale") is not None

    def check(self) -> CheckResult:
        v = self.get_version()
        if v is None:
            return CheckResult(NAME, reason="tailscale not installed")
        extra = {"version": v} if v else {}
        active = self.get_service_active()
        if active is False:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="tailscale installed but tailscaled service is "
                "not active",
                extra_info=extra or None,
            )
        state = self.get_backend_state()
        if state is not None:
            extra["backend_state"] = state
            if state != "Running":
                return CheckResult(
                    NAME,
                    health=HealthStateType.UNHEALTHY,
                    reason="tailscaled service is active but tailscale "
                    f"is not running (state: {state})",
                    extra_info=extra,
                )
        return CheckResult(
            NAME,
            reason=f"tailscale installed ({v or 'version unknown'})"
            + (f", backend {state}" if state else ""),
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return TailscaleComponent(inst)
