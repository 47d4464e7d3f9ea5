"""accelerator-amd-processes — compute processes per GPU.

Reference: components/accelerator/nvidia/processes
(nvmlDeviceGetComputeRunningProcesses — processes/component.go:61).
AMD-first: amdsmi per-GPU process list with VRAM usage and CU occupancy.
"""

from __future__ import annotations

from typing import Callable

from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges
from .shared import SmiComponentMixin

NAME = "accelerator-amd-processes"


class ProcessesComponent(TickerComponent, SmiComponentMixin):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._smi = inst.smi
        self._shared = inst.shared_snapshots
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        # test seam: devices getter
        self.get_devices: Callable = (
            self._smi.devices if self._smi is not None else dict
        )

    def _process_lists(self):
        def fetch():
            out = {}
            for uuid, dev in self.get_devices().items():
                try:
                    out[uuid] = dev.process_list()
                except Exception:
                    out[uuid] = []
            return out

        if self._shared is not None:
            return self._shared.get_aux("process_list", fetch)
        return fetch()

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
        total = 0
        extra = {}
        for uuid, procs in self._process_lists().items():
            total += len(procs)
            self._gauges.set(
                "accelerator_amd_processes_count",
                "Number of compute processes on the GPU",
                len(procs),
                uuid=uuid,
            )
            if procs:
                extra[f"{uuid}.pids"] = ",".join(str(p.get("pid", "?")) for p in procs[:16])
        return CheckResult(
            NAME,
            reason=f"{total} compute process(es) across GPUs",
            extra_info=extra or None,
        )


def new(inst: GPUdInstance) -> Component:
    return ProcessesComponent(inst)
