"""cpu — aggregate usage and load averages.

Reference: components/cpu (gopsutil usage % + load avg — cpu/component.go:28).
"""

from __future__ import annotations

import os

import psutil

from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "cpu"


class CPUComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        psutil.cpu_percent(interval=None)  # prime

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def check(self) -> CheckResult:
        usage = psutil.cpu_percent(interval=None)
        load1, load5, load15 = os.getloadavg()
        cores = psutil.cpu_count() or 1
        self._gauges.set("cpu_usage_percent", "Aggregate CPU usage percent", usage)
        self._gauges.set("cpu_load_average_1min", "1-minute load average", load1)
        self._gauges.set("cpu_load_average_5min", "5-minute load average", load5)
        self._gauges.set("cpu_load_average_15min", "15-minute load average", load15)
        extra = {
            "usage_percent": f"{usage:.1f}",
            "load_1m": f"{load1:.2f}",
            "cores": str(cores),
        }
        return CheckResult(
            NAME,
            reason=f"cpu usage {usage:.1f}%, load1 {load1:.2f} on {cores} cores",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return CPUComponent(inst)
