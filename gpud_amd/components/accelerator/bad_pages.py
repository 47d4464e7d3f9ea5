"""accelerator-amd-bad-pages — retired HBM pages (remapped-rows analog).

Reference: components/accelerator/nvidia/remapped-rows (correctable/
uncorrectable remaps, pending ⇒ reboot, failed ⇒ HW inspection —
remapped-rows/component.go:40-77). AMD-first: amdgpu RAS bad-page
retirement — ``pending`` pages need a reboot to be reserved,
``unreservable`` pages or a count at the EEPROM threshold need hardware
service.
"""

from __future__ import annotations

from typing import Callable

from ...apiv1.types import (
    HealthStateType,
    RepairActionType,
    SuggestedActions,
)
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-bad-pages"


class BadPagesComponent(TickerComponent, SmiComponentMixin):
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
        pending, unreservable, threshold_hit = [], [], []
        extra = {}
        for uuid, snap in snaps.items():
            bp = snap.get("bad_pages")
            if not bp:
                continue
            total = int(bp.get("total", 0))
            pend = int(bp.get("pending", 0))
            unres = int(bp.get("unreservable", 0))
            thr = int(bp.get("threshold", 0))
            self._gauges.set(
                "accelerator_amd_bad_pages_total",
                "Total retired HBM pages",
                total,
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_bad_pages_pending",
                "Bad pages pending reservation (reboot required)",
                pend,
                uuid=uuid,
            )
            extra[f"{uuid}.bad_pages"] = f"total={total},pending={pend},unres={unres}"
            if pend > 0:
                pending.append(uuid)
            if unres > 0:
                unreservable.append(uuid)
            if thr > 0 and total >= thr:
                threshold_hit.append(uuid)
        if unreservable or threshold_hit:
            bad = sorted(set(unreservable + threshold_hit))
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=(
                    "unreservable bad pages or retirement threshold reached on "
                    + ", ".join(bad)
                ),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="HBM page retirement exhausted",
                    repair_actions=[RepairActionType.HARDWARE_INSPECTION],
                ),
            )
        if pending:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="bad pages pending reservation on " + ", ".join(pending),
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="pending HBM page retirement needs a reboot",
                    repair_actions=[RepairActionType.REBOOT_SYSTEM],
                ),
            )
        return CheckResult(
            NAME,
            reason=f"no pending/unreservable bad pages on {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return BadPagesComponent(inst)
