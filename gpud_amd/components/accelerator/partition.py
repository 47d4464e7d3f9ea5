"""accelerator-amd-partition — compute/memory partition mode (SPX/NPS).

No reference analog (gpud does not monitor NVML MIG); this is
MI355X-specific coverage: CDNA partitioning (SPX/DPX/TPX/QPX/CPX compute
modes × NPS1/NPS4 memory modes) changes how many amdsmi processors
enumerate and how jobs should be scheduled, so a node silently left in
CPX after maintenance is a real operational hazard. Monitor-only: gpud
never calls the partition setters (amdsmi_set_gpu_compute_partition needs
an idle device and, for memory, a driver reload).

Health rules: informational by default; if ``expected_compute_partition``
or ``expected_memory_partition`` is configured (flag or control-plane
updateConfig) a mismatch is Unhealthy with a HW-inspection-free
suggested action (the fix is an operator mode switch, not a repair).
"""

from __future__ import annotations

import time
from typing import Callable, Dict

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-partition"

_PARTITION_TYPES = {"SPX": 1, "DPX": 2, "TPX": 3, "QPX": 4, "CPX": 8}


class PartitionComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        # read through to the live Config each check so control-plane
        # updateConfig takes effect without a restart
        self._cfg = inst.config
        # partition mode only changes via operator action + device quiesce;
        # a 60 s TTL detects that hazard without adding its 3 amdsmi reads
        # per GPU to every poll cycle (measured: +0.8 ms on the 2.0 ms p50)
        self._cache: Dict[str, Dict] = {}
        self._cached_at: float = 0.0
        self.cache_ttl_seconds: float = 60.0
        self.get_partition_info: Callable[[], Dict[str, Dict]] = (
            self._read_partitions
        )

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def is_supported(self) -> bool:
        return self._smi is not None and self._smi.exists

    def _read_partitions(self) -> Dict[str, Dict]:
        """uuid -> partition_info dict, TTL-cached (see __init__)."""
        now = time.monotonic()
        if self._cache and now - self._cached_at < self.cache_ttl_seconds:
            return self._cache
        out: Dict[str, Dict] = {}
        for uuid, dev in self._smi.devices().items():
            try:
                out[uuid] = dev.partition_info()
            except Exception:
                out[uuid] = {}
        self._cache = out
        self._cached_at = now
        return out

    @property
    def _expected_compute(self) -> str:
        return (getattr(self._cfg, "expected_compute_partition", "") or "").upper()

    @property
    def _expected_memory(self) -> str:
        return (getattr(self._cfg, "expected_memory_partition", "") or "").upper()

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        infos = self.get_partition_info()
        extra: Dict[str, str] = {}
        mismatched = []
        modes = set()
        for uuid, info in infos.items():
            if not info:
                continue
            comp = str(info.get("compute_partition", "")).upper()
            mem = str(info.get("memory_partition", "")).upper()
            modes.add(f"{comp or '?'}/{mem or '?'}")
            extra[f"{uuid}.partition"] = f"compute={comp},memory={mem}"
            if comp in _PARTITION_TYPES:
                self._gauges.set(
                    "accelerator_amd_partition_count",
                    "Partitions per OAM implied by the compute partition mode",
                    float(
                        int(info.get("num_partitions", 0))
                        or _PARTITION_TYPES[comp]
                    ),
                    uuid=uuid,
                )
            if self._expected_compute and comp and comp != self._expected_compute:
                mismatched.append(f"{uuid}: compute {comp}")
            if self._expected_memory and mem and mem != self._expected_memory:
                mismatched.append(f"{uuid}: memory {mem}")
        if mismatched:
            want = "/".join(
                x for x in (self._expected_compute, self._expected_memory) if x
            )
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=(
                    f"partition mode mismatch (expected {want}): "
                    + "; ".join(mismatched[:8])
                ),
                extra_info=extra,
            )
        if not modes:
            return CheckResult(NAME, reason="partition mode not reported")
        return CheckResult(
            NAME,
            reason=f"partition mode {', '.join(sorted(modes))} on "
            f"{len(infos)} GPU(s)",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return PartitionComponent(inst)
