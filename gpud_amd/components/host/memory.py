"""memory — host RAM usage.

Reference: components/memory (gopsutil virtual memory — memory/component.go:28).
"""

from __future__ import annotations

import psutil

from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "memory"


class MemoryComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def check(self) -> CheckResult:
        vm = psutil.virtual_memory()
        self._gauges.set("memory_total_bytes", "Total RAM", vm.total)
        self._gauges.set("memory_used_bytes", "Used RAM", vm.used)
        self._gauges.set("memory_available_bytes", "Available RAM", vm.available)
        self._gauges.set("memory_used_percent", "Used RAM percent", vm.percent)
        return CheckResult(
            NAME,
            reason=f"memory {vm.percent:.1f}% used "
            f"({vm.used >> 30} GiB of {vm.total >> 30} GiB)",
            extra_info={
                "total_bytes": str(vm.total),
                "used_percent": f"{vm.percent:.1f}",
            },
        )


def new(inst: GPUdInstance) -> Component:
    return MemoryComponent(inst)
