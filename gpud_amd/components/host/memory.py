"""memory — host RAM usage + OOM/EDAC kernel events.

Reference: components/memory (gopsutil virtual memory — memory/component.go:28;
kmsg matcher for oom / oom_cgroup / edac events — memory testdata fixtures).
"""

from __future__ import annotations

import datetime
import re
from typing import Optional

import psutil

from ...pkg.kmsg.syncer import MatchResult, Syncer
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "memory"

_OOM_RULES = (
    ("memory_oom", re.compile(r"Out of memory: Killed process"), "Warning"),
    ("memory_oom_cgroup", re.compile(r"Memory cgroup out of memory"), "Warning"),
    ("memory_oom_kill_constraint", re.compile(r"oom-kill:constraint="), "Warning"),
    ("memory_edac_correctable", re.compile(r"EDAC MC\d+: \d+ CE"), "Warning"),
    ("memory_edac_uncorrectable", re.compile(r"EDAC MC\d+: \d+ UE"), "Critical"),
)


def match_memory_kmsg(line: str) -> Optional[MatchResult]:
    for name, rx, event_type in _OOM_RULES:
        if rx.search(line):
            return MatchResult(name=name, event_type=event_type, message=line)
    return None


class MemoryComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._kmsg = inst.kmsg_reader
        self._syncer: Optional[Syncer] = None

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def start(self) -> None:
        if self._kmsg is not None and self._bucket is not None:
            self._syncer = Syncer(self._kmsg, match_memory_kmsg, self._bucket)
        super().start()

    def events(self, since: datetime.datetime):
        return self._bucket.get(since) if self._bucket is not None else []

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
