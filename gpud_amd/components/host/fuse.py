"""fuse — FUSE connection congestion.

Reference: components/fuse (/sys/fs/fuse/connections waiting vs
max-background thresholds — fuse/component.go:28-42).
"""

from __future__ import annotations

import os
from typing import Callable, List, Tuple

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "fuse"
CONNECTIONS_DIR = "/sys/fs/fuse/connections"
DEFAULT_CONGESTED_PCT = 90.0


def read_connections(root: str = CONNECTIONS_DIR) -> List[Tuple[str, int, int]]:
    """Returns (conn_id, waiting, max_background) triples."""
    out = []
    if not os.path.isdir(root):
        return out
    for cid in sorted(os.listdir(root)):
        d = os.path.join(root, cid)
        try:
            with open(os.path.join(d, "waiting")) as f:
                waiting = int(f.read().strip() or 0)
            with open(os.path.join(d, "max_background")) as f:
                maxbg = int(f.read().strip() or 0)
        except (OSError, ValueError):
            continue
        out.append((cid, waiting, maxbg))
    return out


class FuseComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        self.connections_dir = CONNECTIONS_DIR
        self.get_connections: Callable = lambda: read_connections(self.connections_dir)

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME]

    def is_supported(self) -> bool:
        return os.path.isdir(self.connections_dir)

    def check(self) -> CheckResult:
        conns = self.get_connections()
        congested = []
        for cid, waiting, maxbg in conns:
            self._gauges.set(
                "fuse_connection_waiting", "FUSE requests waiting", waiting,
                connection=cid,
            )
            if maxbg > 0 and waiting >= maxbg * DEFAULT_CONGESTED_PCT / 100.0:
                congested.append(cid)
        if congested:
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=f"congested FUSE connection(s): {', '.join(congested)}",
            )
        return CheckResult(NAME, reason=f"{len(conns)} FUSE connection(s), none congested")


def new(inst: GPUdInstance) -> Component:
    return FuseComponent(inst)
