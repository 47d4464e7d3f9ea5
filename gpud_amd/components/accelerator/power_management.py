"""accelerator-amd-power-management — GPU power management state.

The persistence-mode analog (reference: components/accelerator/nvidia/
persistence-mode — persistence-mode/component.go:27): on NVIDIA the check
is "is the driver kept initialized"; the AMD equivalent concern is whether
the amdgpu runtime power management is in its expected state
(amdsmi_is_gpu_power_management_enabled). Compute nodes want PM active so
idle GPUs clock down; the check is informational (Degraded on mismatch,
never Unhealthy).
"""

from __future__ import annotations

from typing import Callable

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-power-management"


class PowerManagementComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
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
        disabled, errors = [], []
        n = 0
        for uuid, dev in self.get_devices().items():
            n += 1
            try:
                enabled = dev.power_management_enabled()
            except Exception:
                errors.append(uuid)
                continue
            self._gauges.set(
                "accelerator_amd_power_management_enabled",
                "1 when GPU power management is enabled",
                1.0 if enabled else 0.0,
                uuid=uuid,
            )
            if not enabled:
                disabled.append(uuid)
        if disabled:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason="power management disabled on " + ", ".join(disabled),
            )
        reason = f"power management enabled on {n - len(errors)} GPU(s)"
        if errors:
            reason += f" (unreadable on {len(errors)})"
        return CheckResult(NAME, reason=reason)


def new(inst: GPUdInstance) -> Component:
    return PowerManagementComponent(inst)
