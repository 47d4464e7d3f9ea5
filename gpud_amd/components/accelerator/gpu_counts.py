"""accelerator-amd-gpu-counts — SMI count vs lspci count vs expected.

Reference: components/accelerator/nvidia/gpu-counts (lspci | grep NVIDIA
vs NVML device count vs expected — gpu-counts/component.go:216-368).
AMD-first: count AMD display/processing accelerators on the PCI bus
(vendor 0x1002, Instinct devices report as "Processing accelerators"),
compare with the amdsmi enumeration and the configured expected count.
"""

from __future__ import annotations

import os

import subprocess
from typing import Callable, Optional

from ...apiv1.types import HealthStateType, RepairActionType, SuggestedActions
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-gpu-counts"


def count_amd_gpus_lspci(lspci_command: str = "") -> Optional[int]:
    """Count AMD accelerator/VGA functions via lspci -n (vendor 1002)."""
    cmd = lspci_command or "lspci"
    try:
        out = subprocess.run(
            [cmd, "-n"], capture_output=True, text=True, timeout=10
        ).stdout
    except (OSError, subprocess.TimeoutExpired):
        return None
    n = 0
    for line in out.splitlines():
        # "0a:00.0 1200: 1002:75a0" — class 0x1200 processing accelerator,
        # 0x0300/0x0380 display classes; count vendor 1002 in those classes
        parts = line.split()
        if len(parts) >= 3 and parts[2].startswith("1002:"):
            cls = parts[1].rstrip(":")
            if cls.startswith(("1200", "0300", "0302", "0380")):
                n += 1
    return n


def count_render_nodes(dri_dir: str = "/dev/dri") -> Optional[int]:
    """amdgpu render nodes (/dev/dri/renderD*) — the DCGM-style /dev
    triangulation source (reference: pkg/nvidia/dev/device_count.go
    counting /dev/nvidia[0-9]+). On partitioned (DPX/QPX/CPX) boards one
    physical GPU exposes several render nodes, so this is compared
    against the amdsmi enumeration (which is also per-partition), not
    the PCI count."""
    try:
        import re as _re

        return sum(
            1 for n in os.listdir(dri_dir) if _re.fullmatch(r"renderD\d+", n)
        )
    except OSError:
        return None


class GPUCountsComponent(TickerComponent, SmiComponentMixin):
    # The PCI topology changes only on hotplug/driver rebind, and a GPU
    # dropping off the driver is caught every cycle by the amdsmi
    # enumeration below — so the lspci crosscheck (a fork+exec) is cached
    # for 5 minutes instead of spawned per check.
    LSPCI_TTL_SECONDS = 300.0

    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self.expected = inst.expected_gpu_count
        self._cfg = inst.config
        self._lspci_command = inst.lspci_command
        self._lspci_cache: Optional[int] = None
        self._lspci_cached_at = 0.0
        self.count_lspci: Callable[[], Optional[int]] = self._cached_lspci
        self.count_render: Callable[[], Optional[int]] = count_render_nodes

    def _cached_lspci(self) -> Optional[int]:
        import time

        from ... import smi as smi_pkg

        if smi_pkg.mock_enabled():
            # mock backend: the real PCI bus is unrelated to the mocked GPUs
            return None
        now = time.monotonic()
        if now - self._lspci_cached_at > self.LSPCI_TTL_SECONDS:
            self._lspci_cache = count_amd_gpus_lspci(self._lspci_command)
            self._lspci_cached_at = now
        return self._lspci_cache

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        return True  # meaningful even when SMI is missing (counts mismatch!)

    def check(self) -> CheckResult:
        # pick up control-plane updateConfig changes live (the shared
        # Config object is mutated in place)
        cfg_expected = (
            getattr(self._cfg, "expected_gpu_count", 0) if self._cfg else 0
        )
        if cfg_expected > 0:
            self.expected = cfg_expected
        smi_count = 0
        smi_err = ""
        if self._smi is not None and self._smi.exists:
            try:
                smi_count = self._smi.device_count()
            except Exception as e:
                smi_err = str(e)
        pci_count = self.count_lspci()
        self._gauges.set(
            "accelerator_amd_gpu_counts_smi",
            "GPUs enumerated by amdsmi",
            smi_count,
        )
        if pci_count is not None:
            self._gauges.set(
                "accelerator_amd_gpu_counts_lspci",
                "AMD accelerators on the PCI bus",
                pci_count,
            )
        render_count = None
        from ... import smi as smi_pkg

        if not smi_pkg.mock_enabled():
            render_count = self.count_render()
            if render_count is not None:
                self._gauges.set(
                    "accelerator_amd_gpu_counts_render_nodes",
                    "amdgpu render nodes under /dev/dri",
                    render_count,
                )
        extra = {
            "smi_count": str(smi_count),
            "lspci_count": "" if pci_count is None else str(pci_count),
            "render_nodes": "" if render_count is None else str(render_count),
            "expected": str(self.expected),
        }
        if smi_err:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="amdsmi device enumeration failed",
                error=smi_err,
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="GPU enumeration failure",
                    repair_actions=[RepairActionType.REBOOT_SYSTEM],
                ),
            )
        if self.expected > 0 and smi_count < self.expected:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"expected {self.expected} GPU(s), amdsmi sees {smi_count}",
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="missing GPU(s)",
                    repair_actions=[
                        RepairActionType.REBOOT_SYSTEM,
                        RepairActionType.HARDWARE_INSPECTION,
                    ],
                ),
            )
        if render_count is not None and smi_count and render_count < smi_count:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=(
                    f"only {render_count} render node(s) under /dev/dri for "
                    f"{smi_count} amdsmi device(s) — device files missing "
                    "(container device-cgroup or udev problem)"
                ),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="GPU device files missing",
                    repair_actions=[RepairActionType.CHECK_USER_APP_AND_GPU],
                ),
            )
        if pci_count is not None and smi_count and pci_count < smi_count:
            # fewer devices on the bus than the driver claims — real trouble
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=(
                    f"lspci sees {pci_count} AMD accelerator(s) but amdsmi "
                    f"enumerates {smi_count} — a GPU may have dropped off the bus"
                ),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="PCI/driver GPU count mismatch",
                    repair_actions=[RepairActionType.REBOOT_SYSTEM],
                ),
            )
        if pci_count is not None and smi_count and pci_count > smi_count:
            # more on the bus than the driver exposes: either a GPU fell off
            # the driver, or this process runs in a container whose device
            # cgroup hides some render nodes while sysfs shows the host bus.
            # Without expected_gpu_count we cannot tell, so flag Degraded;
            # set expected_gpu_count for a hard rule.
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=(
                    f"lspci sees {pci_count} AMD accelerator(s) but amdsmi "
                    f"exposes {smi_count} — GPU missing from the driver, or "
                    "container device visibility is restricted"
                ),
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"gpu counts consistent (smi={smi_count}"
            + (f", lspci={pci_count}" if pci_count is not None else "")
            + ")",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return GPUCountsComponent(inst)
