"""memory — host RAM usage + vmalloc/BPF-JIT buffers + OOM/EDAC events.

Reference: components/memory (gopsutil virtual memory — memory/component.go:28;
VmallocTotal/VmallocUsed from the same snapshot — component.go:192-193;
BPF JIT buffer bytes summed from /proc/vmallocinfo — bpf.go:22-42, the
early-warning signal for the bpf_jit_limit exhaustion failure mode that
breaks container networking; kmsg matcher for oom / oom_cgroup / edac
events — memory testdata fixtures).
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


def read_bpf_jit_buffer_bytes(path: str = "/proc/vmallocinfo") -> Optional[int]:
    """Sum of bpf_jit allocation sizes (reference: memory/bpf.go:40-60 —
    `grep bpf_jit /proc/vmallocinfo | awk '{s+=$2}'`). Needs root; returns
    None when the file is unreadable (non-root, masked procfs)."""
    try:
        total = 0
        with open(path) as f:
            for ln in f:
                if "bpf_jit" not in ln:
                    continue
                parts = ln.split()
                if len(parts) >= 2:
                    try:
                        total += int(parts[1])
                    except ValueError:
                        continue
        return total
    except OSError:
        return None


def read_vmalloc_meminfo(path: str = "/proc/meminfo"):
    """(VmallocTotal, VmallocUsed) in bytes, or (None, None)."""
    total = used = None
    try:
        with open(path) as f:
            for ln in f:
                if ln.startswith("VmallocTotal:"):
                    total = int(ln.split()[1]) * 1024
                elif ln.startswith("VmallocUsed:"):
                    used = int(ln.split()[1]) * 1024
    except OSError:
        pass
    return total, used


class MemoryComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self._bucket = (
            inst.event_store.bucket(NAME) if inst.event_store is not None else None
        )
        self._kmsg = inst.kmsg_reader
        self._syncer: Optional[Syncer] = None
        # injectable proc paths (reference test seam: bpf.go file param)
        self.vmallocinfo_path = "/proc/vmallocinfo"
        self.meminfo_path = "/proc/meminfo"

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
        extra = {
            "total_bytes": str(vm.total),
            "used_percent": f"{vm.percent:.1f}",
        }
        vm_total, vm_used = read_vmalloc_meminfo(self.meminfo_path)
        if vm_total is not None:
            extra["vmalloc_total_bytes"] = str(vm_total)
        if vm_used is not None:
            extra["vmalloc_used_bytes"] = str(vm_used)
            self._gauges.set(
                "memory_vmalloc_used_bytes", "VmallocUsed", vm_used
            )
        bpf = read_bpf_jit_buffer_bytes(self.vmallocinfo_path)
        if bpf is not None:
            extra["bpf_jit_buffer_bytes"] = str(bpf)
            self._gauges.set(
                "memory_bpf_jit_buffer_bytes",
                "Sum of bpf_jit allocations in /proc/vmallocinfo",
                bpf,
            )
        return CheckResult(
            NAME,
            reason=f"memory {vm.percent:.1f}% used "
            f"({vm.used >> 30} GiB of {vm.total >> 30} GiB)",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return MemoryComponent(inst)
