"""accelerator-amd-clock-speed — per-GPU gfx/memory clocks.

Reference: components/accelerator/nvidia/clock-speed (graphics + mem clock
gauges — clock_speed.go:34-41). AMD-first: amdsmi GFX and MEM clock domains
with their maxima; deep-sleep state is reported but never unhealthy.
"""

from __future__ import annotations

from typing import Callable

from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-clock-speed"


class ClockSpeedComponent(TickerComponent, SmiComponentMixin):
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
            c = snap.get("clock")
            if not c:
                continue
            gfx = float(c.get("gfx_mhz", 0))
            mem = float(c.get("mem_mhz", 0))
            self._gauges.set(
                "accelerator_amd_clock_speed_gfx_mhz",
                "Current GFX (shader) clock",
                gfx,
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_clock_speed_mem_mhz",
                "Current memory (HBM) clock",
                mem,
                uuid=uuid,
            )
            extra[f"{uuid}.gfx_mhz"] = str(int(gfx))
            # deep-sleep is informational — an idle GPU legitimately clocks
            # down (never unhealthy; reference clock-speed is also
            # info-only)
            if c.get("gfx_deep_sleep"):
                extra[f"{uuid}.gfx_deep_sleep"] = "1"
        return CheckResult(
            NAME,
            reason=f"clock speeds collected for {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return ClockSpeedComponent(inst)
