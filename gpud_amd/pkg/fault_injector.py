"""Fault injector (reference: pkg/fault-injector/fault_injector.go:12-67).

Two injection surfaces, same as the reference:

1. kmsg injection — write a synthetic kernel message into /dev/kmsg so the
   live RAS watcher (and any dmesg consumer) picks it up; known catalog
   entries can be injected by name (reference Xid id → canned NVRM line).
2. SMI-level injection — ``SMIFailureInjector`` flags consumed by the smi
   mock/wrapper layer (reference: components/registry.go:82-109
   FailureInjector: GPU-lost, requires-reset, bad-page pending, throttle,
   xGMI-unhealthy, device-enumeration error).
"""

from __future__ import annotations

from dataclasses import dataclass, field
from typing import Dict, Optional, Set

from . import ras_catalog
from .kmsg.writer import Writer


@dataclass
class KernelMessage:
    message: str = ""
    priority: int = 2


@dataclass
class Request:
    """A fault-injection request (reference fault_injector.go Request)."""

    ras_event_name: str = ""  # catalog entry name → canned message
    kernel_message: Optional[KernelMessage] = None

    @staticmethod
    def from_dict(d: Dict) -> "Request":
        km = d.get("kernel_message") or d.get("kernelMessage")
        return Request(
            ras_event_name=d.get("ras_event_name", "") or d.get("rasEventName", ""),
            kernel_message=KernelMessage(
                message=km.get("message", ""), priority=int(km.get("priority", 2))
            )
            if km
            else None,
        )


class Injector:
    def __init__(self, kmsg_writer: Writer):
        self._writer = kmsg_writer

    def kmsg_writer(self) -> Writer:
        return self._writer

    def inject(self, req: Request) -> Optional[str]:
        """Returns an error string, or None on success."""
        if req.ras_event_name:
            msg = ras_catalog.get_message_to_inject(req.ras_event_name)
            if msg is None:
                return f"unknown injectable RAS event {req.ras_event_name!r}"
            return self._writer.write(msg, priority=2)
        if req.kernel_message is not None and req.kernel_message.message:
            return self._writer.write(
                req.kernel_message.message, priority=req.kernel_message.priority
            )
        return "empty fault-injection request"


@dataclass
class SMIFailureInjector:
    """SMI-level failure flags (reference: components/registry.go:82-109).

    UUIDs listed here make the smi layer report the corresponding failure,
    letting the daemon be exercised end-to-end on healthy hardware.
    """

    gpu_lost_uuids: Set[str] = field(default_factory=set)
    requires_reset_uuids: Set[str] = field(default_factory=set)
    bad_page_pending_uuids: Set[str] = field(default_factory=set)
    bad_page_threshold_uuids: Set[str] = field(default_factory=set)
    throttle_uuids: Set[str] = field(default_factory=set)
    thermal_throttle_uuids: Set[str] = field(default_factory=set)
    xgmi_unhealthy_uuids: Set[str] = field(default_factory=set)
    ecc_uncorrectable_uuids: Set[str] = field(default_factory=set)
    product_name_override: str = ""
    device_enumeration_error: str = ""

    def any_active(self) -> bool:
        return bool(
            self.gpu_lost_uuids
            or self.requires_reset_uuids
            or self.bad_page_pending_uuids
            or self.bad_page_threshold_uuids
            or self.throttle_uuids
            or self.thermal_throttle_uuids
            or self.xgmi_unhealthy_uuids
            or self.ecc_uncorrectable_uuids
            or self.product_name_override
            or self.device_enumeration_error
        )
