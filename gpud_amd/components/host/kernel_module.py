"""kernel-module — required kernel modules loaded?

Reference: components/kernel-module (GPUdInstance.KernelModulesToCheck vs
loaded modules — kernel-module/component.go:21,30-44). On an MI355X node
the interesting defaults are ``amdgpu`` and ``amdttm``/``amdkcl`` variants;
the list is configuration-driven like the reference.
"""

from __future__ import annotations

from typing import Callable, List, Set

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "kernel-module"


def loaded_modules(proc_modules: str = "/proc/modules") -> Set[str]:
    mods: Set[str] = set()
    try:
        with open(proc_modules) as f:
            for line in f:
                name = line.split(" ", 1)[0].strip()
                if name:
                    mods.add(name)
    except OSError:
        pass
    return mods


class KernelModuleComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self.modules_to_check: List[str] = list(inst.kernel_modules_to_check or [])
        self.get_loaded: Callable = loaded_modules

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def check(self) -> CheckResult:
        if not self.modules_to_check:
            return CheckResult(NAME, reason="no kernel modules configured to check")
        loaded = self.get_loaded()
        missing = [m for m in self.modules_to_check if m not in loaded]
        if missing:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="missing kernel module(s): " + ", ".join(missing),
                extra_info={"missing": ",".join(missing)},
            )
        return CheckResult(
            NAME,
            reason=f"all {len(self.modules_to_check)} required module(s) loaded",
        )


def new(inst: GPUdInstance) -> Component:
    return KernelModuleComponent(inst)
