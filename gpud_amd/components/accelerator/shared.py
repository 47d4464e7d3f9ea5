"""Shared per-cycle telemetry snapshot for all accelerator components.

The reference's poll model collects data once per interval and lets
components share data sources (reference: docs/ARCHITECTURE.md:3-5; one
nvml.Instance shared through GPUdInstance — pkg/server/server.go:277-328).
We go one step further for flat 8-GPU overhead: ONE native
``metrics_snapshot_all()`` sweep per poll cycle feeds every accelerator
component through this TTL cache, so a full accelerator scan costs one
GIL-released C++ sweep instead of ~14 independent SMI walks.
"""

from __future__ import annotations

import threading
import time
from typing import Any, Dict, Optional

DEFAULT_TTL_SECONDS = 10.0


class SharedSnapshots:
    def __init__(self, smi_instance: Any, ttl_seconds: float = DEFAULT_TTL_SECONDS):
        self.smi = smi_instance
        self.ttl = ttl_seconds
        self._lock = threading.Lock()
        self._snapshots: Dict[str, Dict[str, Any]] = {}
        self._taken_at: float = 0.0
        # per-cycle memo for heavier SMI calls several components need
        # (process_list is used by both processes and gpm; link_metrics by
        # xgmi) — cleared on every refresh so data stays cycle-fresh
        self._aux: Dict[str, Any] = {}

    def get(self, max_age: Optional[float] = None) -> Dict[str, Dict[str, Any]]:
        """Snapshots keyed by uuid, refreshed when older than the TTL."""
        ttl = self.ttl if max_age is None else max_age
        with self._lock:
            now = time.monotonic()
            if now - self._taken_at > ttl:
                self._snapshots = self.smi.snapshot_all()
                self._aux = {}
                self._taken_at = time.monotonic()
            return self._snapshots

    def refresh(self) -> Dict[str, Dict[str, Any]]:
        with self._lock:
            self._snapshots = self.smi.snapshot_all()
            self._aux = {}
            self._taken_at = time.monotonic()
            return self._snapshots

    def get_aux(self, key: str, fn) -> Any:
        """Memoize ``fn()`` for the current poll cycle under ``key``."""
        with self._lock:
            if key in self._aux:
                return self._aux[key]
        val = fn()
        with self._lock:
            self._aux[key] = val
        return val


class SmiComponentMixin:
    """Guard chain shared by accelerator components.

    Mirrors the reference's per-check guard chain (reference:
    components/accelerator/nvidia/temperature/component.go:129-157):
    nil instance / not exists / init error ⇒ healthy no-op states with an
    explanatory reason, so the daemon runs cleanly on GPU-less hosts.
    """

    def smi_guard(self):
        """Returns a CheckResult to short-circuit with, or None to proceed.

        Cached once the instance is known-good: existence cannot regress at
        runtime (GPU-lost injection filters devices(), not the session), so
        the healthy path costs one attribute read per check."""
        if getattr(self, "_smi_guard_ok", False):
            return None
        from ..base import CheckResult
        from ...apiv1.types import HealthStateType

        inst = getattr(self, "_smi", None)
        if inst is None:
            return CheckResult(
                self.name,
                health=HealthStateType.HEALTHY,
                reason="no SMI instance (no AMD GPU on this host)",
            )
        if not inst.exists:
            err = inst.init_error()
            return CheckResult(
                self.name,
                health=HealthStateType.HEALTHY,
                reason="amdsmi library not loaded (no AMD GPU driver)",
                error=err if err else "",
            )
        self._smi_guard_ok = True
        return None
