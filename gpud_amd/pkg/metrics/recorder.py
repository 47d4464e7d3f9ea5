"""Self-telemetry recorder (reference: pkg/metrics/recorder/gpud_metrics.go).

Periodically records the daemon's own footprint: CPU%, RSS, open FDs, state
DB size, cumulative SQLite op latencies — the instruments behind the
daemon-overhead baseline. Also owns the per-component
``check_duration_seconds`` histogram the reference lacks; TickerComponent
feeds it via ``observe_check_duration``.
"""

from __future__ import annotations

import threading
from typing import Optional

from prometheus_client import CollectorRegistry, Gauge, Histogram

from ..log import logger
from ..sqlite_util import Conn, read_db_size
from .registry import LABEL_COMPONENT

try:
    import psutil
except ImportError:  # pragma: no cover
    psutil = None


class Recorder:
    def __init__(
        self,
        registry: CollectorRegistry,
        db_rw: Optional[Conn] = None,
        interval_seconds: float = 900.0,  # 15 min, like the reference
    ):
        self._db_rw = db_rw
        self._interval = interval_seconds
        self._stop = threading.Event()
        self._proc = psutil.Process() if psutil else None
        if self._proc is not None:
            self._proc.cpu_percent(interval=None)  # prime the sampler

        self.fd_usage = Gauge(
            "gpud_file_descriptor_usage",
            "Number of open file descriptors held by gpud",
            [LABEL_COMPONENT],
            registry=registry,
        ).labels(**{LABEL_COMPONENT: "gpud"})
        self.cpu_percent = Gauge(
            "gpud_cpu_percent",
            "gpud process CPU usage percent",
            [LABEL_COMPONENT],
            registry=registry,
        ).labels(**{LABEL_COMPONENT: "gpud"})
        self.rss_bytes = Gauge(
            "gpud_resident_memory_bytes",
            "gpud process resident set size in bytes",
            [LABEL_COMPONENT],
            registry=registry,
        ).labels(**{LABEL_COMPONENT: "gpud"})
        self.db_size = Gauge(
            "gpud_state_db_size_bytes",
            "Size of the gpud state SQLite database in bytes",
            [LABEL_COMPONENT],
            registry=registry,
        ).labels(**{LABEL_COMPONENT: "gpud"})
        self.sqlite_insert_seconds = Gauge(
            "gpud_sqlite_insert_update_total_seconds",
            "Cumulative seconds spent in SQLite insert/update ops",
            [LABEL_COMPONENT],
            registry=registry,
        ).labels(**{LABEL_COMPONENT: "gpud"})
        self.sqlite_select_seconds = Gauge(
            "gpud_sqlite_select_total_seconds",
            "Cumulative seconds spent in SQLite select ops",
            [LABEL_COMPONENT],
            registry=registry,
        ).labels(**{LABEL_COMPONENT: "gpud"})
        self.check_duration = Histogram(
            "gpud_component_check_duration_seconds",
            "Duration of one component Check() call",
            [LABEL_COMPONENT],
            registry=registry,
            buckets=(
                0.0005, 0.001, 0.0025, 0.005, 0.01, 0.025, 0.05, 0.1,
                0.25, 0.5, 1.0, 2.5, 5.0, 10.0, 30.0,
            ),
        )

        self._duration_children: dict = {}
        self._thread = threading.Thread(
            target=self._run, daemon=True, name="gpud-recorder"
        )

    def observe_check_duration(self, component: str, seconds: float) -> None:
        child = self._duration_children.get(component)
        if child is None:
            child = self.check_duration.labels(**{LABEL_COMPONENT: component})
            self._duration_children[component] = child
        child.observe(seconds)

    def record_once(self) -> None:
        if self._proc is not None:
            try:
                self.fd_usage.set(self._proc.num_fds())
                self.cpu_percent.set(self._proc.cpu_percent(interval=None))
                self.rss_bytes.set(self._proc.memory_info().rss)
            except Exception:
                logger.exception("self-telemetry process sampling failed")
        if self._db_rw is not None:
            try:
                self.db_size.set(read_db_size(self._db_rw))
                self.sqlite_insert_seconds.set(self._db_rw.total_insert_seconds)
                self.sqlite_select_seconds.set(self._db_rw.total_select_seconds)
            except Exception:
                logger.exception("self-telemetry DB sampling failed")

    def start(self) -> None:
        self._thread.start()

    def _run(self) -> None:
        self.record_once()
        while not self._stop.wait(self._interval):
            try:
                self.record_once()
            except Exception:
                logger.exception("self-telemetry record failed")

    def stop(self) -> None:
        self._stop.set()
