"""Scrape→persist→purge loop (reference: pkg/metrics/syncer/syncer.go:22-84).

Runs every minute by default (wired like pkg/server/server.go:231-242):
scrape the Prometheus registry, persist to SQLite, purge rows older than the
retention period.
"""

from __future__ import annotations

import datetime
import threading

from ..log import logger
from .scraper import Scraper
from .store import MetricsStore


class Syncer:
    def __init__(
        self,
        scraper: Scraper,
        store: MetricsStore,
        sync_interval_seconds: float = 60.0,
        retention: datetime.timedelta = datetime.timedelta(days=3),
    ):
        self._scraper = scraper
        self._store = store
        self._interval = sync_interval_seconds
        self._retention = retention
        self._stop = threading.Event()
        self._thread: threading.Thread = threading.Thread(
            target=self._run, daemon=True, name="gpud-metrics-syncer"
        )

    def start(self) -> None:
        self._thread.start()

    def sync_once(self) -> int:
        metrics = self._scraper.scrape()
        self._store.record(metrics)
        cutoff = datetime.datetime.now(datetime.timezone.utc) - self._retention
        self._store.purge(cutoff)
        return len(metrics)

    def _run(self) -> None:
        while not self._stop.wait(self._interval):
            try:
                self.sync_once()
            except Exception:
                logger.exception("metrics sync failed")

    def stop(self) -> None:
        self._stop.set()
