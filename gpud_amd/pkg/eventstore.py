"""Per-component SQLite event store.

Schema-compatible with the reference (reference: pkg/eventstore/database.go):
one table per component bucket named ``components_<name>_events_v0_5_0``
(name sanitized: spaces/dashes → underscores, lower-cased — database.go:137),
columns ``(timestamp INTEGER, name TEXT, type TEXT, message TEXT,
extra_info TEXT)`` with indexes on timestamp/name/type, and a periodic purge
honouring the retention period (default 3 days — pkg/eventstore/types.go:53).
"""

from __future__ import annotations

import datetime
import json
import threading
import time
from typing import Dict, List, Optional

from ..apiv1.types import Event
from .log import logger
from .sqlite_util import Conn

SCHEMA_VERSION = "v0_5_0"
DEFAULT_RETENTION = datetime.timedelta(days=3)


def default_table_name(component_name: str, schema_version: str = SCHEMA_VERSION) -> str:
    c = component_name.replace(" ", "_").replace("-", "_")
    while "__" in c:
        c = c.replace("__", "_")
    return f"components_{c.lower()}_events_{schema_version}"


class Bucket:
    """One component's event table (reference: eventstore table struct)."""

    def __init__(
        self,
        db_rw: Conn,
        db_ro: Conn,
        component_name: str,
        retention: datetime.timedelta = DEFAULT_RETENTION,
        disable_purge: bool = False,
    ):
        self.component_name = component_name
        self.table = default_table_name(component_name)
        self._db_rw = db_rw
        self._db_ro = db_ro
        self.retention = retention if not disable_purge else datetime.timedelta(0)
        self._stop = threading.Event()
        self._purge_thread: Optional[threading.Thread] = None
        self._create_table()
        if self.retention.total_seconds() > 1 and not disable_purge:
            interval = max(self.retention.total_seconds() / 5, 1.0)
            self._purge_thread = threading.Thread(
                target=self._run_purge, args=(interval,), daemon=True,
                name=f"gpud-purge-{component_name}",
            )
            self._purge_thread.start()

    def _create_table(self) -> None:
        t = self.table
        # the v0_4_0-era table is unused — dropped at open like the
        # reference (eventstore/database.go:97-103), so a gpud.state file
        # carried over from an old reference install opens cleanly
        legacy = default_table_name(self.component_name, "v0_4_0")
        self._db_rw.executescript(f"DROP TABLE IF EXISTS {legacy};")
        self._db_rw.executescript(
            f"""
CREATE TABLE IF NOT EXISTS {t} (
    timestamp INTEGER NOT NULL,
    name TEXT NOT NULL,
    type TEXT NOT NULL,
    message TEXT,
    extra_info TEXT
);
CREATE INDEX IF NOT EXISTS idx_{t}_timestamp ON {t}(timestamp);
CREATE INDEX IF NOT EXISTS idx_{t}_name ON {t}(name);
CREATE INDEX IF NOT EXISTS idx_{t}_type ON {t}(type);
"""
        )

    # -- write path ---------------------------------------------------------

    def insert(self, ev: Event, extra_info: Optional[Dict[str, str]] = None) -> None:
        self._db_rw.execute(
            f"INSERT INTO {self.table} (timestamp, name, type, message, extra_info)"
            " VALUES (?, ?, ?, ?, ?)",
            (
                int(ev.time.timestamp()),
                ev.name,
                ev.type,
                ev.message,
                json.dumps(extra_info, sort_keys=True) if extra_info else None,
            ),
        )

    # -- read path ----------------------------------------------------------

    def _row_to_event(self, row) -> Event:
        ts, name, typ, message, _extra = row
        return Event(
            time=datetime.datetime.fromtimestamp(ts, tz=datetime.timezone.utc),
            component=self.component_name,
            name=name,
            type=typ,
            message=message or "",
        )

    def get(self, since: datetime.datetime) -> List[Event]:
        """Events at/after ``since``, newest first (reference Get semantics)."""
        rows = self._db_ro.query(
            f"SELECT timestamp, name, type, message, extra_info FROM {self.table}"
            " WHERE timestamp >= ? ORDER BY timestamp DESC",
            (int(since.timestamp()),),
        )
        return [self._row_to_event(r) for r in rows]

    def find(self, ev: Event) -> Optional[Event]:
        """Find an identical (timestamp, name, type) event — dedup helper."""
        row = self._db_ro.query_one(
            f"SELECT timestamp, name, type, message, extra_info FROM {self.table}"
            " WHERE timestamp = ? AND name = ? AND type = ? LIMIT 1",
            (int(ev.time.timestamp()), ev.name, ev.type),
        )
        return self._row_to_event(row) if row else None

    def find_by_name_since(self, name: str, since: datetime.datetime) -> List[Event]:
        rows = self._db_ro.query(
            f"SELECT timestamp, name, type, message, extra_info FROM {self.table}"
            " WHERE name = ? AND timestamp >= ? ORDER BY timestamp DESC",
            (name, int(since.timestamp())),
        )
        return [self._row_to_event(r) for r in rows]

    def latest(self) -> Optional[Event]:
        row = self._db_ro.query_one(
            f"SELECT timestamp, name, type, message, extra_info FROM {self.table}"
            " ORDER BY timestamp DESC LIMIT 1"
        )
        return self._row_to_event(row) if row else None

    # -- purge --------------------------------------------------------------

    def purge(self, before_unix: int) -> int:
        n = self._db_ro.query_one(
            f"SELECT COUNT(*) FROM {self.table} WHERE timestamp < ?", (before_unix,)
        )
        self._db_rw.execute(
            f"DELETE FROM {self.table} WHERE timestamp < ?", (before_unix,)
        )
        return int(n[0]) if n else 0

    def _run_purge(self, interval: float) -> None:
        while not self._stop.wait(interval):
            try:
                cutoff = int(time.time() - self.retention.total_seconds())
                purged = self.purge(cutoff)
                if purged:
                    logger.info("purged %d events from %s", purged, self.table)
            except Exception:
                logger.exception("event purge failed for %s", self.table)

    def close(self) -> None:
        self._stop.set()


class Store:
    """Reference: pkg/eventstore Store — a Bucket factory sharing the DB pair."""

    def __init__(
        self,
        db_rw: Conn,
        db_ro: Conn,
        retention: datetime.timedelta = DEFAULT_RETENTION,
    ):
        self._db_rw = db_rw
        self._db_ro = db_ro
        self.retention = retention
        self._buckets: Dict[str, Bucket] = {}
        self._lock = threading.Lock()

    def bucket(self, name: str, disable_purge: bool = False) -> Bucket:
        with self._lock:
            b = self._buckets.get(name)
            if b is None:
                b = Bucket(
                    self._db_rw,
                    self._db_ro,
                    name,
                    retention=self.retention,
                    disable_purge=disable_purge,
                )
                self._buckets[name] = b
            return b

    def close(self) -> None:
        with self._lock:
            for b in self._buckets.values():
                b.close()
            self._buckets.clear()
