"""pci — ACS (Access Control Services) status on PCI bridges.

Reference: components/pci (lspci ACS status on bridges; ACS-on hurts GPU
P2P — pci/component.go:26). On MI355X nodes ACS on the bridges above the
GPUs forces P2P/xGMI-adjacent PCIe DMA through the root complex.
The check is virtualization-gated like the reference
(pci/component.go:159-168): inside KVM guests the hypervisor owns the
topology and ACS state is expected, so the check is skipped there.
"""

from __future__ import annotations

import shutil
import subprocess
from typing import Callable, List, Optional

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "pci"


def parse_acs_bridges(lspci_vvv_output: str) -> List[str]:
    """Bridges whose ACS control has SrcValid enabled, from `lspci -vvv`
    text (split out for fixture-driven tests)."""
    enabled = []
    current_dev = ""
    is_bridge = False
    in_acs_cap = False
    for line in lspci_vvv_output.splitlines():
        if line and not line[0].isspace():
            current_dev = line.split(" ", 1)[0]
            is_bridge = "PCI bridge" in line
            in_acs_cap = False
            continue
        if not is_bridge:
            continue
        s = line.strip()
        if s.startswith("Capabilities:") and "Access Control Services" in s:
            in_acs_cap = True
            continue
        if in_acs_cap and s.startswith("ACSCtl:"):
            # e.g. "ACSCtl: SrcValid+ TransBlk- ReqRedir+ ..."
            if "SrcValid+" in s:
                enabled.append(current_dev)
            in_acs_cap = False
    return enabled


def bridges_with_acs_enabled(lspci_command: str = "") -> Optional[List[str]]:
    cmd = lspci_command or "lspci"
    try:
        out = subprocess.run(
            [cmd, "-vvv"], capture_output=True, text=True, timeout=30
        )
        if out.returncode != 0:
            return None
    except (OSError, subprocess.TimeoutExpired):
        return None
    return parse_acs_bridges(out.stdout)


def detect_virt_env() -> str:
    """`systemd-detect-virt --vm` output: "none" on bare metal (and in
    plain containers on bare metal), "kvm"/"qemu"/"vmware"/... inside VM
    guests, "" when undeterminable (reference:
    pkg/host/virtualization_environment.go:21 — the VM field drives the
    ACS skip, a container runtime alone does not)."""
    path = shutil.which("systemd-detect-virt")
    if path is None:
        return ""
    try:
        out = subprocess.run(
            [path, "--vm"], capture_output=True, text=True, timeout=10
        )
        # exit code 1 means "none" — stdout still carries the answer
        return (out.stdout or "").strip()
    except (OSError, subprocess.TimeoutExpired):
        return ""


class PCIComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._lspci_command = inst.lspci_command
        self.get_acs_bridges: Callable = lambda: bridges_with_acs_enabled(
            self._lspci_command
        )
        self.get_virt_env: Callable = detect_virt_env

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def check(self) -> CheckResult:
        virt = self.get_virt_env()
        if virt and virt != "none":
            return CheckResult(
                NAME,
                reason=f"host virt env is {virt} (no need to check ACS)",
                extra_info={"virt_env": virt},
            )
        bridges = self.get_acs_bridges()
        if bridges is None:
            return CheckResult(NAME, reason="lspci unavailable; ACS check skipped")
        if bridges:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=f"ACS enabled on {len(bridges)} PCI bridge(s) — "
                "disable ACS for full GPU P2P bandwidth",
                extra_info={"bridges": ",".join(bridges[:16])},
            )
        return CheckResult(NAME, reason="no PCI bridges with ACS enabled")


def new(inst: GPUdInstance) -> Component:
    return PCIComponent(inst)
