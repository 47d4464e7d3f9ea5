"""accelerator-amd-temperature — per-GPU edge/hotspot/HBM temperatures.

Reference: components/accelerator/nvidia/temperature (current/limit/margin
temps, margin-threshold Degraded state — component.go:118-180, threshold.go).
AMD-first: amdsmi edge/hotspot(junction)/VRAM sensors with their CRITICAL
limits; Degraded when the margin to the hotspot limit falls below the
configured threshold, Unhealthy at/over the limit.
"""

from __future__ import annotations

from typing import Callable

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-temperature"

DEFAULT_MARGIN_THRESHOLD_C = 10.0


class TemperatureComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        cfg = inst.config
        self.margin_threshold_c = getattr(
            cfg, "temperature_margin_threshold_c", DEFAULT_MARGIN_THRESHOLD_C
        ) if cfg else DEFAULT_MARGIN_THRESHOLD_C
        # test seam (reference pattern: injected getter function fields)
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
        unhealthy, degraded = [], []
        extra = {}
        for uuid, snap in snaps.items():
            t = snap.get("temperature")
            if not t:
                continue
            hotspot = float(t.get("hotspot_c", 0))
            limit = float(t.get("hotspot_limit_c", 0))
            self._gauges.set(
                "accelerator_amd_temperature_edge_celsius",
                "Current GPU edge temperature",
                float(t.get("edge_c", 0)),
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_temperature_hotspot_celsius",
                "Current GPU hotspot (junction) temperature",
                hotspot,
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_temperature_vram_celsius",
                "Current HBM temperature",
                float(t.get("vram_c", 0)),
                uuid=uuid,
            )
            if limit > 0:
                margin = limit - hotspot
                self._gauges.set(
                    "accelerator_amd_temperature_hotspot_margin_celsius",
                    "Margin between hotspot temperature and its critical limit",
                    margin,
                    uuid=uuid,
                )
                extra[f"{uuid}.hotspot_c"] = str(int(hotspot))
                if margin <= 0:
                    unhealthy.append(uuid)
                elif margin < self.margin_threshold_c:
                    degraded.append(uuid)
        if unhealthy:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"hotspot temperature at/over critical limit on {', '.join(unhealthy)}",
                extra_info=extra,
            )
        if degraded:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=(
                    f"hotspot margin below {self.margin_threshold_c:.0f}C on "
                    + ", ".join(degraded)
                ),
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"all {len(snaps)} GPU(s) within temperature limits",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return TemperatureComponent(inst)
