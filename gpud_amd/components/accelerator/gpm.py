"""accelerator-amd-gpm — fine-grained GPU performance metrics.

Reference: components/accelerator/nvidia/gpm (NVML GPM dual-sample deltas:
SM occupancy, tensor/fp utilization — gpm/component.go:34-43). AMD-first:
amdsmi gpu_metrics current clocks + engine activity plus per-process CU
occupancy, which is the CDNA-native occupancy signal (256 CUs per MI355X).
"""

from __future__ import annotations

from typing import Callable

from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-gpm"


class GPMComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self.get_snapshots: Callable = (
            self._shared.get if self._shared is not None else lambda: {}
        )
        self.get_devices: Callable = (
            self._smi.devices if self._smi is not None else dict
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
            gm = snap.get("gpu_metrics")
            if gm:
                self._gauges.set(
                    "accelerator_amd_gpm_current_gfxclk_mhz",
                    "Instant GFX clock from gpu_metrics",
                    float(gm.get("current_gfxclk_mhz", 0)),
                    uuid=uuid,
                )
                self._gauges.set(
                    "accelerator_amd_gpm_current_uclk_mhz",
                    "Instant memory clock from gpu_metrics",
                    float(gm.get("current_uclk_mhz", 0)),
                    uuid=uuid,
                )
        # per-process CU occupancy (the SM-occupancy analog on 256-CU CDNA4);
        # process lists are memoized per cycle in the shared cache
        try:
            def fetch():
                out = {}
                for uuid, dev in self.get_devices().items():
                    try:
                        out[uuid] = dev.process_list()
                    except Exception:
                        out[uuid] = []
                return out

            plists = (
                self._shared.get_aux("process_list", fetch)
                if self._shared is not None
                else fetch()
            )
            for uuid, procs in plists.items():
                occ = sum(int(p.get("cu_occupancy", 0)) for p in procs)
                self._gauges.set(
                    "accelerator_amd_gpm_cu_occupancy",
                    "Summed CU occupancy of compute processes (of 256 CUs)",
                    occ,
                    uuid=uuid,
                )
                extra[f"{uuid}.cu_occupancy"] = str(occ)
        except Exception:
            pass
        return CheckResult(
            NAME,
            reason=f"performance metrics collected for {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return GPMComponent(inst)
