"""accelerator-amd-bad-envs — dangerous globally-set ROCm/RCCL env vars.

Reference analog: accelerator-nvidia-bad-envs (docs/COMPONENTS.md — tracks
bad environment variables globally set for the GPUs; since removed from
the reference tree but the hazard is real). AMD-first curation: variables
that silently serialize, hide devices, or cripple the fabric when an
operator leaves them set system-wide after debugging:

- ``AMD_SERIALIZE_KERNEL`` / ``AMD_SERIALIZE_COPY`` — debug serialization,
  order-of-magnitude slowdowns;
- ``HIP_LAUNCH_BLOCKING`` / ``CUDA_LAUNCH_BLOCKING`` — synchronous
  launches;
- ``HSA_OVERRIDE_GFX_VERSION`` — forces the wrong ISA (wrong code objects
  on MI355X);
- ``ROCR_VISIBLE_DEVICES`` / ``HIP_VISIBLE_DEVICES`` /
  ``CUDA_VISIBLE_DEVICES`` — device hiding has no business being global;
- ``NCCL_P2P_DISABLE`` / ``RCCL_P2P_DISABLE`` — forces collectives off the
  xGMI mesh;
- ``HSA_ENABLE_SDMA=0`` — disables the copy engines;
- ``GPU_MAX_HW_QUEUES=1`` — single hardware queue.

Scanned scopes: /etc/environment and PID 1's environment (the "global"
scopes services inherit), plus the daemon's own environment. Any hit ⇒
Degraded with the offending scope/variable named.
"""

from __future__ import annotations

import os
import time
from typing import Callable, Dict, List, Tuple

from ...apiv1.types import HealthStateType
from ..base import CheckResult, Component, GPUdInstance, TickerComponent
from ..metrics_util import ComponentGauges

NAME = "accelerator-amd-bad-envs"

# var -> predicate on its value (None = any value is bad)
BAD_ENV_RULES: Dict[str, object] = {
    "AMD_SERIALIZE_KERNEL": None,
    "AMD_SERIALIZE_COPY": None,
    "HIP_LAUNCH_BLOCKING": lambda v: v not in ("", "0"),
    "CUDA_LAUNCH_BLOCKING": lambda v: v not in ("", "0"),
    "HSA_OVERRIDE_GFX_VERSION": None,
    "ROCR_VISIBLE_DEVICES": None,
    "HIP_VISIBLE_DEVICES": None,
    "CUDA_VISIBLE_DEVICES": None,
    "NCCL_P2P_DISABLE": lambda v: v not in ("", "0"),
    "RCCL_P2P_DISABLE": lambda v: v not in ("", "0"),
    "HSA_ENABLE_SDMA": lambda v: v == "0",
    "GPU_MAX_HW_QUEUES": lambda v: v.isdigit() and int(v) <= 1,
}


def _is_bad(var: str, value: str) -> bool:
    rule = BAD_ENV_RULES.get(var)
    if rule is None and var in BAD_ENV_RULES:
        return True
    if callable(rule):
        try:
            return bool(rule(value))
        except Exception:  # noqa: BLE001 — a weird value is still a finding
            return True
    return False


def read_etc_environment(path: str = "/etc/environment") -> Dict[str, str]:
    out: Dict[str, str] = {}
    try:
        with open(path) as f:
            for line in f:
                line = line.strip()
                if not line or line.startswith("#") or "=" not in line:
                    continue
                k, v = line.split("=", 1)
                out[k.strip()] = v.strip().strip('"').strip("'")
    except OSError:
        pass
    return out


def read_pid1_environ(path: str = "/proc/1/environ") -> Dict[str, str]:
    out: Dict[str, str] = {}
    try:
        with open(path, "rb") as f:
            for chunk in f.read().split(b"\0"):
                if b"=" in chunk:
                    k, v = chunk.split(b"=", 1)
                    out[k.decode("utf-8", "replace")] = v.decode(
                        "utf-8", "replace"
                    )
    except OSError:
        pass  # needs root; absent scope is simply not scanned
    return out


# device-hiding vars are legitimate per-process (launchers set them); they
# are only a finding in the truly global scopes
_GLOBAL_ONLY = {"ROCR_VISIBLE_DEVICES", "HIP_VISIBLE_DEVICES",
                "CUDA_VISIBLE_DEVICES"}
_GLOBAL_SCOPES = {"/etc/environment", "pid1"}


def scan_bad_envs(
    scopes: Dict[str, Dict[str, str]]
) -> List[Tuple[str, str, str]]:
    """[(scope, var, value)] for every bad variable in every scope."""
    findings = []
    for scope, env in scopes.items():
        for var, value in env.items():
            if var in _GLOBAL_ONLY and scope not in _GLOBAL_SCOPES:
                continue
            if _is_bad(var, value):
                findings.append((scope, var, value))
    return findings


class BadEnvsComponent(TickerComponent):
    def __init__(self, inst: GPUdInstance):
        super().__init__()
        self._gauges = ComponentGauges(NAME, inst.metrics_registry)
        # the global scopes change only via operator action; re-read them
        # on a 60 s TTL and keep the daemon's own env live
        self._global_cache: Dict[str, Dict[str, str]] = {}
        self._cached_at = 0.0
        self.cache_ttl_seconds = 60.0
        self.get_scopes: Callable[[], Dict[str, Dict[str, str]]] = (
            self._default_scopes
        )

    def _default_scopes(self) -> Dict[str, Dict[str, str]]:
        now = time.monotonic()
        if not self._global_cache or now - self._cached_at > self.cache_ttl_seconds:
            self._global_cache = {
                "/etc/environment": read_etc_environment(),
                "pid1": read_pid1_environ(),
            }
            self._cached_at = now
        return {**self._global_cache, "daemon": dict(os.environ)}

    @property
    def name(self) -> str:
        return NAME

    def tags(self) -> list:
        return ["accelerator", "amd", "gpu", NAME]

    def check(self) -> CheckResult:
        findings = scan_bad_envs(self.get_scopes())
        self._gauges.set(
            "accelerator_amd_bad_envs_found",
            "Count of dangerous globally-set ROCm/RCCL environment variables",
            float(len(findings)),
        )
        if findings:
            shown = "; ".join(
                f"{scope}: {var}={value!r}" for scope, var, value in findings[:8]
            )
            return CheckResult(
                NAME,
                health=HealthStateType.DEGRADED,
                reason=f"dangerous GPU env var(s) set globally — {shown}",
                extra_info={
                    f"{scope}.{var}": value for scope, var, value in findings[:16]
                },
            )
        return CheckResult(NAME, reason="no bad GPU environment variables set")


def new(inst: GPUdInstance) -> Component:
    return BadEnvsComponent(inst)
