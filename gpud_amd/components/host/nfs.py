"""nfs — NFS member-visibility checker.

Reference: components/nfs + pkg/nfs-checker (write/read a per-machine file
in each configured NFS group dir, verify member visibility; configs pushed
from the control plane — nfs/component.go:31).
"""

from __future__ import annotations

import os
from dataclasses import dataclass
from typing import List

from ...apiv1.types import HealthStateType
from ...pkg import host as pkghost
from ..base import CheckResult, Component, GPUdInstance, TickerComponent

NAME = "nfs"


@dataclass
class GroupConfig:
    """One NFS group dir every member machine writes its marker into."""

    volume_path: str
    dir_name: str = ".gpud-nfs-checker"
    file_contents: str = ""


class NFSComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self.configs: List[GroupConfig] = [
            c if isinstance(c, GroupConfig) else GroupConfig(**c)
            for c in (inst.nfs_checker_configs or [])
        ]
        self.machine_id = pkghost.machine_id() or pkghost.hostname()

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def set_configs(self, configs: List[GroupConfig]) -> None:
        """Control-plane push path (reference: configs pushed via session)."""
        self.configs = configs

    def check(self) -> CheckResult:
        if not self.configs:
            return CheckResult(NAME, reason="no NFS group configs")
        problems = []
        extra = {}
        for cfg in self.configs:
            d = os.path.join(cfg.volume_path, cfg.dir_name)
            marker = os.path.join(d, self.machine_id)
            try:
                os.makedirs(d, exist_ok=True)
                contents = cfg.file_contents or self.machine_id
                with open(marker, "w") as f:
                    f.write(contents)
                with open(marker) as f:
                    back = f.read()
                if back != contents:
                    problems.append(f"{cfg.volume_path}: read-back mismatch")
                    continue
                members = [
                    n for n in os.listdir(d) if not n.startswith(".")
                ]
                extra[cfg.volume_path] = f"{len(members)} member(s) visible"
            except OSError as e:
                problems.append(f"{cfg.volume_path}: {e}")
        if problems:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="; ".join(problems),
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"all {len(self.configs)} NFS group dir(s) writable and readable",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return NFSComponent(inst)
