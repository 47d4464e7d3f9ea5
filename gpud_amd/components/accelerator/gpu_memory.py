"""accelerator-amd-memory — per-GPU HBM usage.

Reference: components/accelerator/nvidia/memory (memory info v2 —
memory.go:65-83). AMD-first: amdsmi VRAM usage against the 288 GB HBM3E
pool per MI355X.
"""

from __future__ import annotations

from typing import Callable

from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-memory"


class GPUMemoryComponent(TickerComponent, SmiComponentMixin):
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
            v = snap.get("vram")
            if not v:
                continue
            total_mb = float(v.get("vram_total_mb", 0))
            used_mb = float(v.get("vram_used_mb", 0))
            self._gauges.set(
                "accelerator_amd_memory_total_bytes",
                "Total HBM capacity",
                total_mb * 1024 * 1024,
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_memory_used_bytes",
                "Used HBM",
                used_mb * 1024 * 1024,
                uuid=uuid,
            )
            if total_mb > 0:
                self._gauges.set(
                    "accelerator_amd_memory_used_percent",
                    "Used HBM percent",
                    100.0 * used_mb / total_mb,
                    uuid=uuid,
                )
            extra[f"{uuid}.vram_used_mb"] = str(int(used_mb))
        return CheckResult(
            NAME,
            reason=f"HBM usage collected for {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return GPUMemoryComponent(inst)
