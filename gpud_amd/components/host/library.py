"""library — required shared libraries resolvable?

Reference: components/library (configured libraries resolvable in search
dirs — library/component.go:102-111). AMD defaults of interest:
libamd_smi.so, librccl.so, libamdhip64.so under /opt/rocm/lib.
"""

from __future__ import annotations

import glob
import os
from typing import Dict, List

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "library"

DEFAULT_SEARCH_DIRS = ["/usr/lib", "/usr/lib64", "/usr/lib/x86_64-linux-gnu", "/opt/rocm/lib"]


def resolve(library: str, search_dirs: List[str]) -> bool:
    for d in search_dirs:
        if glob.glob(os.path.join(d, library)) or glob.glob(
            os.path.join(d, library + "*")
        ):
            return True
    return False


class LibraryComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        # mapping: library glob -> extra search dirs
        self.libraries: Dict[str, List[str]] = dict(inst.libraries_to_check or {})

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def check(self) -> CheckResult:
        if not self.libraries:
            return CheckResult(NAME, reason="no libraries configured to check")
        missing = []
        for lib, dirs in self.libraries.items():
            search = (dirs or []) + DEFAULT_SEARCH_DIRS
            if not resolve(lib, search):
                missing.append(lib)
        if missing:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="unresolvable libraries: " + ", ".join(missing),
            )
        return CheckResult(NAME, reason=f"all {len(self.libraries)} libraries resolvable")


def new(inst: GPUdInstance) -> Component:
    return LibraryComponent(inst)
