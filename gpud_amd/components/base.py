"""Component model: the contract every health-check component implements.

Mirrors the reference's ``components/types.go:20-110`` (Component interface,
CheckResult, optional Deregisterable/HealthSettable interfaces) and
``components/registry.go`` (Registry, GPUdInstance DI container), re-designed
for Python: components are objects with a ``check()`` method driven by a
per-component ticker thread; the last check result is cached under a lock and
served to the HTTP layer without re-polling (reference poll model:
docs/ARCHITECTURE.md — collect once per interval, reads come from cache).
"""

from __future__ import annotations

import abc
import datetime
import threading
import time
from dataclasses import dataclass, field
from typing import Any, Callable, Dict, List, Optional

from ..apiv1.types import (
    Event,
    HealthState,
    HealthStateType,
    RunModeType,
    SuggestedActions,
    utcnow,
)
from ..pkg.log import logger

# Default per-component poll interval — same cadence as the reference
# (reference: components/accelerator/nvidia/temperature/component.go:83).
DEFAULT_POLL_INTERVAL_SECONDS = 60.0


class CheckResult:
    """Concrete result of one ``Component.check()``.

    Reference: components/types.go CheckResult interface — we flatten it to a
    data-carrying class; components may subclass to add typed payload fields.
    """

    def __init__(
        self,
        component_name: str,
        health: str = HealthStateType.HEALTHY,
        reason: str = "",
        error: str = "",
        suggested_actions: Optional[SuggestedActions] = None,
        extra_info: Optional[Dict[str, str]] = None,
        run_mode: str = "",
        component_type: str = "",
        raw_output: str = "",
        ts: Optional[datetime.datetime] = None,
    ):
        self.component_name = component_name
        self.health = health
        self.reason = reason
        self.error = error
        self.suggested_actions = suggested_actions
        self.extra_info = extra_info
        self.run_mode = run_mode
        self.component_type = component_type
        self.raw_output = raw_output
        self.ts = ts or utcnow()

    def summary(self) -> str:
        return self.reason

    def health_state_type(self) -> str:
        return self.health

    def health_states(self) -> List[HealthState]:
        return [
            HealthState(
                time=self.ts,
                component=self.component_name,
                component_type=self.component_type,
                name=self.component_name,
                run_mode=self.run_mode,
                health=self.health,
                reason=self.reason,
                error=self.error,
                suggested_actions=self.suggested_actions,
                extra_info=self.extra_info,
                raw_output=self.raw_output,
            )
        ]

    def __str__(self) -> str:
        return f"{self.component_name}: {self.health} ({self.reason})"


class Component(abc.ABC):
    """Reference: components/types.go:20 Component interface (8 methods)."""

    @property
    @abc.abstractmethod
    def name(self) -> str:
        ...

    def tags(self) -> List[str]:
        return []

    def is_supported(self) -> bool:
        return True

    @abc.abstractmethod
    def start(self) -> None:
        ...

    @abc.abstractmethod
    def check(self) -> CheckResult:
        ...

    @abc.abstractmethod
    def last_health_states(self) -> List[HealthState]:
        ...

    def events(self, since: datetime.datetime) -> List[Event]:
        return []

    @abc.abstractmethod
    def close(self) -> None:
        ...

    # -- optional capabilities (reference: components/types.go:70-110) ------

    def deregisterable(self) -> bool:
        return False

    def can_set_healthy(self) -> bool:
        return False

    def set_healthy(self) -> None:  # pragma: no cover - optional capability
        raise NotImplementedError


class TickerComponent(Component):
    """Base class running ``check()`` on a background ticker thread.

    Equivalent of the per-component goroutine loop in the reference
    (reference: components/accelerator/nvidia/temperature/component.go:81-97):
    check once immediately on start, then on every tick; cache the last
    result under a lock; record the check duration into the metrics registry
    (the reference lacks a per-check duration histogram — we add one, it is
    the instrument behind the poll-latency baseline).
    """

    #: override per subclass
    poll_interval: float = DEFAULT_POLL_INTERVAL_SECONDS
    run_mode: str = RunModeType.AUTO

    def __init__(self) -> None:
        self._lock = threading.Lock()
        self._last_check_result: Optional[CheckResult] = None
        self._stop = threading.Event()
        self._thread: Optional[threading.Thread] = None
        self._duration_observer: Optional[Callable[[str, float], None]] = None

    def set_duration_observer(self, fn: Callable[[str, float], None]) -> None:
        self._duration_observer = fn

    def start(self) -> None:
        if self.run_mode == RunModeType.MANUAL:
            return
        if self._thread is not None:
            return
        self._thread = threading.Thread(
            target=self._run_loop, name=f"gpud-{self.name}", daemon=True
        )
        self._thread.start()

    def _run_loop(self) -> None:
        while not self._stop.is_set():
            self.trigger_check()
            if self._stop.wait(self.poll_interval):
                return

    def trigger_check(self) -> CheckResult:
        """Run one check now, cache the result, record its duration."""
        t0 = time.monotonic()
        try:
            cr = self.check()
        except Exception as e:  # a component must never kill the daemon
            logger.exception("component %s check failed", self.name)
            cr = CheckResult(
                component_name=self.name,
                health=HealthStateType.UNHEALTHY,
                reason="component check raised an exception",
                error=str(e),
            )
        dur = time.monotonic() - t0
        if self._duration_observer is not None:
            try:
                self._duration_observer(self.name, dur)
            except Exception:
                pass
        with self._lock:
            self._last_check_result = cr
        return cr

    def last_health_states(self) -> List[HealthState]:
        with self._lock:
            cr = self._last_check_result
        if cr is None:
            return [
                HealthState(
                    component=self.name,
                    name=self.name,
                    health=HealthStateType.INITIALIZING,
                    reason="check not yet run",
                    run_mode=self.run_mode,
                )
            ]
        return cr.health_states()

    def last_check_result(self) -> Optional[CheckResult]:
        with self._lock:
            return self._last_check_result

    def close(self) -> None:
        self._stop.set()
        t = self._thread
        if t is not None and t.is_alive() and t is not threading.current_thread():
            t.join(timeout=2.0)


@dataclass
class GPUdInstance:
    """Dependency-injection container shared by every component.

    Reference: components/registry.go:24-109 GPUdInstance. One shared SMI
    handle for all accelerator components (never one init per component —
    SURVEY.md §7 hard parts), shared stores, command overrides that double as
    test seams, and the failure injector.
    """

    smi: Any = None  # gpud_amd.smi.Instance (or None on GPU-less hosts)
    shared_snapshots: Any = None  # accelerator.shared.SharedSnapshots
    db_rw: Any = None  # sqlite3 connection pool (read-write)
    db_ro: Any = None  # sqlite3 connection pool (read-only)
    event_store: Any = None  # pkg.eventstore.Store
    reboot_event_store: Any = None  # pkg.host.RebootEventStore
    metrics_registry: Any = None  # prometheus CollectorRegistry
    kmsg_reader: Any = None  # pkg.kmsg.Reader (shared /dev/kmsg access)
    mount_points: List[str] = field(default_factory=list)
    mount_targets: List[str] = field(default_factory=list)
    kernel_modules_to_check: List[str] = field(default_factory=list)
    libraries_to_check: Dict[str, List[str]] = field(default_factory=dict)
    nfs_checker_configs: List[Any] = field(default_factory=list)
    expected_gpu_count: int = 0
    # nsenter-style command overrides (test seams; reference registry.go:46-78)
    reboot_command: str = ""
    findmnt_command: str = ""
    lsblk_command: str = ""
    df_command: str = ""
    lspci_command: str = ""
    containerd_address: str = ""
    # failure injection (reference: components/registry.go:82-109)
    failure_injector: Any = None
    # health thresholds pushed by flags / control plane
    config: Any = None


class Registry:
    """Thread-safe component registry.

    Reference: components/registry.go:112-231 — MustRegister/Register/All/
    Get/Deregister. ``all_components`` preserves registration order (the
    reference keeps a sorted list; we keep insertion order, which is the
    canonical registration order from components/all/all.go).
    """

    def __init__(self, gpud_instance: Optional[GPUdInstance] = None):
        self._lock = threading.Lock()
        self._components: Dict[str, Component] = {}
        self._order: List[str] = []
        self.gpud_instance = gpud_instance or GPUdInstance()

    def register(self, init_fn: Callable[[GPUdInstance], Component]) -> Component:
        c = init_fn(self.gpud_instance)
        with self._lock:
            if c.name in self._components:
                raise ValueError(f"component {c.name!r} already registered")
            self._components[c.name] = c
            self._order.append(c.name)
        return c

    def must_register(self, init_fn: Callable[[GPUdInstance], Component]) -> Component:
        return self.register(init_fn)

    def register_component(self, c: Component) -> Component:
        with self._lock:
            if c.name in self._components:
                raise ValueError(f"component {c.name!r} already registered")
            self._components[c.name] = c
            self._order.append(c.name)
        return c

    def all_components(self) -> List[Component]:
        with self._lock:
            return [self._components[n] for n in self._order]

    def get(self, name: str) -> Optional[Component]:
        with self._lock:
            return self._components.get(name)

    def deregister(self, name: str) -> Optional[Component]:
        with self._lock:
            c = self._components.get(name)
            if c is None:
                return None
            if not c.deregisterable():
                return None
            del self._components[name]
            self._order.remove(name)
        return c

    def names(self) -> List[str]:
        with self._lock:
            return list(self._order)
