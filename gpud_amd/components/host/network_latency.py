"""network-latency — edge reachability / latency probe.

Reference: components/network-latency (DERP latency to global edge via
tailscale netcheck; unhealthy iff ALL regions exceed the threshold —
network/latency/component.go:42-44). Re-designed without the tailscale
dependency: TCP connect latency to a configurable target list; with no
targets configured the check is a healthy no-op (air-gapped clusters).
"""

from __future__ import annotations

import socket
import time
from typing import Callable, Dict, List, Tuple

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "network-latency"

DEFAULT_THRESHOLD_MS = 1000.0


def tcp_latency_ms(host: str, port: int, timeout: float = 3.0) -> float:
    """Returns connect latency in ms, or -1 on failure."""
    t0 = time.monotonic()
    try:
        with socket.create_connection((host, port), timeout=timeout):
            return (time.monotonic() - t0) * 1000.0
    except OSError:
        return -1.0


class NetworkLatencyComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        cfg = inst.config
        self.targets: List[Tuple[str, int]] = list(
            getattr(cfg, "latency_targets", []) or []
        ) if cfg else []
        self.threshold_ms = DEFAULT_THRESHOLD_MS
        self.probe: Callable = tcp_latency_ms

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return [NAME, "network"]

    def check(self) -> CheckResult:
        if not self.targets:
            return CheckResult(NAME, reason="no latency targets configured")
        results: Dict[str, float] = {}
        for host, port in self.targets:
            ms = self.probe(host, port)
            results[f"{host}:{port}"] = ms
            if ms >= 0:
                self._gauges.set(
                    "network_latency_connect_milliseconds",
                    "TCP connect latency",
                    ms,
                    target=f"{host}:{port}",
                )
        ok = {k: v for k, v in results.items() if 0 <= v <= self.threshold_ms}
        extra = {k: f"{v:.1f}ms" if v >= 0 else "unreachable" for k, v in results.items()}
        if not ok:
            return CheckResult(
                NAME,
                health=HealthStateType.UNHEALTHY,
                reason="all latency targets unreachable or over "
                f"{self.threshold_ms:.0f}ms",
                extra_info=extra,
            )
        return CheckResult(
            NAME,
            reason=f"{len(ok)}/{len(results)} targets within threshold",
            extra_info=extra,
        )


def new(inst: GPUdInstance) -> Component:
    return NetworkLatencyComponent(inst)
