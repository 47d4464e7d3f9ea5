"""accelerator-amd-ecc — RAS ECC error counters.

Reference: components/accelerator/nvidia/ecc (volatile+aggregate
corrected/uncorrected, unhealthy on uncorrected — ecc/component.go:55,220).
AMD-first: amdsmi total ECC counts (correctable / uncorrectable / deferred
across RAS blocks). Uncorrectable > 0 ⇒ Unhealthy with reboot + hardware
inspection; a rising correctable rate is surfaced as an event.
"""

from __future__ import annotations

import datetime
from typing import Callable, Dict

from ...apiv1.types import (
    Event,
    EventType,
    HealthStateType,
    RepairActionType,
    SuggestedActions,
    utcnow,
)
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-ecc"


class ECCComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._last_correctable: Dict[str, int] = {}
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

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

    # RAS block bits (amdsmi_gpu_block_t): queried only when UEs appear
    _BLOCKS = (
        ("UMC", 1 << 0),
        ("SDMA", 1 << 1),
        ("GFX", 1 << 2),
        ("MMHUB", 1 << 3),
        ("XGMI_WAFL", 1 << 7),
    )

    def _blame_blocks(self, uuid: str) -> str:
        try:
            dev = self._smi.devices().get(uuid)
            if dev is None:
                return ""
            hits = []
            for name, bit in self._BLOCKS:
                try:
                    ec = dev.ecc_count_block(bit)
                except Exception:
                    continue
                ue = int(ec.get("uncorrectable", 0))
                if ue > 0:
                    hits.append(f"{name}:{ue}")
            return ",".join(hits)
        except Exception:
            return ""

    def check(self) -> CheckResult:
        guard = self.smi_guard()
        if guard is not None:
            return guard
        snaps = self.get_snapshots()
        unhealthy = []
        extra = {}
        for uuid, snap in snaps.items():
            e = snap.get("ecc")
            if not e:
                continue
            corr = int(e.get("correctable", 0))
            uncorr = int(e.get("uncorrectable", 0))
            deferred = int(e.get("deferred", 0))
            self._gauges.set(
                "accelerator_amd_ecc_correctable_total",
                "Total correctable ECC errors",
                corr,
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_ecc_uncorrectable_total",
                "Total uncorrectable ECC errors",
                uncorr,
                uuid=uuid,
            )
            self._gauges.set(
                "accelerator_amd_ecc_deferred_total",
                "Total deferred ECC errors",
                deferred,
                uuid=uuid,
            )
            extra[f"{uuid}.ecc"] = f"ce={corr},ue={uncorr},de={deferred}"
            if uncorr > 0:
                unhealthy.append((uuid, uncorr))
                # attribute the UE to RAS blocks (per-block queries are
                # extra ioctls, so they run only on the failure path)
                blocks = self._blame_blocks(uuid)
                if blocks:
                    extra[f"{uuid}.ue_blocks"] = blocks
            prev = self._last_correctable.get(uuid)
            if prev is not None and corr > prev and self._bucket is not None:
                self._bucket.insert(
                    Event(
                        time=utcnow(),
                        component=NAME,
                        name="amd_ecc_correctable_increase",
                        type=EventType.WARNING,
                        message=f"correctable ECC count on {uuid} rose {prev}→{corr}",
                    )
                )
            self._last_correctable[uuid] = corr
        if unhealthy:
            ids = ", ".join(f"{u} (ue={n})" for u, n in unhealthy)
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason=f"uncorrectable ECC errors detected: {ids}",
                extra_info=extra,
                suggested_actions=SuggestedActions(
                    description="uncorrectable ECC (HBM) errors",
                    repair_actions=[
                        RepairActionType.REBOOT_SYSTEM,
                        RepairActionType.HARDWARE_INSPECTION,
                    ],
                ),
            )
        return CheckResult(
            NAME,
            reason=f"no uncorrectable ECC errors on {len(snaps)} GPU(s)",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return ECCComponent(inst)
