"""tailscale — tailscaled presence/version.

Reference: components/tailscale (tailscale/component.go:20).
"""

from __future__ import annotations

import shutil
import subprocess
from typing import Callable, Optional

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


class TailscaleComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self.get_version: Callable = tailscale_version

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
        return CheckResult(
            NAME,
            reason=f"tailscale installed ({v or 'version unknown'})",
            extra_info={"version": v} if v else None,
        )


def new(inst: GPUdInstance) -> Component:
    return TailscaleComponent(inst)
