"""accelerator-amd-utilization — per-GPU engine activity.

Reference: components/accelerator/nvidia/utilization (gpu%/mem% —
utilization.go:36). AMD-first: amdsmi GFX / UMC (memory controller) / MM
engine activity percentages.
"""

from __future__ import annotations

from typing import Callable

from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-utilization"


class UtilizationComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self.get_snapshots: Callable = (
            self._shared.get if self._shared is not None else lambda: {}
        )

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        return self._smi is not None and self._smi.exists

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        snaps = self.get_snapshots()
        extra = {}
        for uuid, snap in snaps.items():
            a = snap.get("activity")
            if not a:
                continue
            self._gauges.set(
                "accelerator_amd_utilization_gfx_percent",
                "GFX engine activity percent",
                float(a.get("gfx_activity_pct", 0)),
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_utilization_umc_percent",
                "Memory-controller (UMC) activity percent",
                float(a.get("umc_activity_pct", 0)),
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_utilization_mm_percent",
                "Multimedia engine activity percent",
                float(a.get("mm_activity_pct", 0)),
                uuid=uuid,
            )
            extra[f"{uuid}.gfx_pct"] = str(a.get("gfx_activity_pct", 0))
        return CheckResult(
            NAME,
            reason=f"utilization collected for {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return UtilizationComponent(inst)
