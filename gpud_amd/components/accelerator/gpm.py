"""accelerator-amd-gpm — fine-grained GPU performance metrics.

Reference: components/accelerator/nvidia/gpm (NVML GPM dual-sample deltas:
SM occupancy, tensor/fp utilization — gpm/component.go:34-43). AMD-first:
amdsmi gpu_metrics current clocks + engine activity plus per-process CU
occupancy, which is the CDNA-native occupancy signal (256 CUs per MI355X).

Per-pipe breakdown on CDNA4: amdsmi does NOT expose an MFMA-pipe
utilization counter (gpu_metrics' ``mm_activity`` is MultiMedia, not
matrix; MFMA busy cycles are rocprof PMCs — SQ_VALU_MFMA_BUSY_CYCLES —
which need a profiling session, not a telemetry ioctl). The per-engine
signals that DO exist and are surfaced here: per-XCC (per-XCD)
instantaneous gfx busy from xcp_stats (8 XCDs per MI355X — an unhealthy
XCD shows as an outlier against its siblings; exported per-XCC plus a
max-min spread gauge) and the gfx/umc/mm activity set. The ACTIVE MFMA
utilization measurement lives in the diag component's MFMA stress kernels
(PMC-verified 97.5% MfmaUtil — profiles/diag_kernels_r1.txt).
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
            act = snap.get("activity") or {}
            xcc = act.get("xcc_busy_pct") or []
            if xcc:
                for i, busy in enumerate(xcc):
                    self._gauges.set(
                        "accelerator_amd_gpm_xcc_busy_percent",
                        "Per-XCC (per-XCD) instantaneous gfx busy",
                        float(busy),
                        uuid=uuid,
                        xcc=str(i),
                    )
                spread = max(xcc) - min(xcc)
                self._gauges.set(
                    "accelerator_amd_gpm_xcc_busy_spread_percent",
                    "max-min spread of per-XCC busy (outlier XCD signal)",
                    float(spread),
                    uuid=uuid,
                )
                extra[f"{uuid}.xcc_busy"] = ",".join(str(b) for b in xcc)
            if "mm_activity_pct" in act:
                self._gauges.set(
                    "accelerator_amd_gpm_mm_activity_percent",
                    "MultiMedia engine activity (NOT matrix/MFMA)",
                    float(act["mm_activity_pct"]),
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
