"""accelerator-amd-power — per-GPU socket power vs enforced limit.

Reference: components/accelerator/nvidia/power (power usage + enforced
limit — power/power.go:39-49). AMD-first: amdsmi socket power, board power
limit and the power-cap; Degraded when sustained at ≥98% of the limit.
"""

from __future__ import annotations

from typing import Callable

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-power"


class PowerComponent(TickerComponent, SmiComponentMixin):
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
        at_limit = []
        extra = {}
        for uuid, snap in snaps.items():
            p = snap.get("power")
            if not p:
                continue
            usage = float(p.get("current_socket_power_w") or p.get("socket_power_w", 0))
            limit = float(p.get("power_limit_w", 0))
            self._gauges.set(
                "accelerator_amd_power_usage_watts",
                "Current GPU socket power draw",
                usage,
                uuid=uuid,
            )
            if limit > 0:
                self._gauges.set(
                    "accelerator_amd_power_limit_watts",
                    "Enforced GPU power limit",
                    limit,
                    uuid=uuid,
                )
                self._gauges.set(
                    "accelerator_amd_power_usage_percent",
                    "Power draw as percent of the enforced limit",
                    100.0 * usage / limit,
                    uuid=uuid,
                )
                extra[f"{uuid}.power_w"] = str(int(usage))
                if usage >= 0.98 * limit:
                    at_limit.append(uuid)
        if at_limit:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason="GPU(s) at power limit (≥98%): " + ", ".join(at_limit),
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"power within limits on {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return PowerComponent(inst)
