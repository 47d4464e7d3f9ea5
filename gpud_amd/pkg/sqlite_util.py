"""SQLite helpers (reference: pkg/sqlite/sqlite.go).

Same pragmas as the reference conn string (sqlite.go:57): WAL journal,
busy_timeout=5000, synchronous=NORMAL; split read-write / read-only
connections; ``compact`` = VACUUM (sqlite.go:123); ``read_db_size`` via
PRAGMA page_count*page_size (sqlite.go:100-111).

Python sqlite3 objects are wrapped in a small lock-guarded ``Conn`` so the
per-component ticker threads and the HTTP serving threads can share them.
"""

from __future__ import annotations

import os
import sqlite3
import threading
import time
from typing import Any, Iterable, List, Optional, Sequence, Tuple


class Conn:
    """A lock-guarded sqlite3 connection usable from any thread."""

    def __init__(self, conn: sqlite3.Connection, readonly: bool = False):
        self._conn = conn
        self._lock = threading.Lock()
        self.readonly = readonly
        # cumulative op latency self-telemetry
        # (reference: pkg/metrics/recorder records sqlite op seconds)
        self.total_insert_seconds = 0.0
        self.total_select_seconds = 0.0

    def execute(self, sql: str, params: Sequence[Any] = ()) -> None:
        t0 = time.monotonic()
        with self._lock:
            self._conn.execute(sql, params)
            self._conn.commit()
        self.total_insert_seconds += time.monotonic() - t0

    def executemany(self, sql: str, rows: Iterable[Sequence[Any]]) -> None:
        t0 = time.monotonic()
        with self._lock:
            self._conn.executemany(sql, rows)
            self._conn.commit()
        self.total_insert_seconds += time.monotonic() - t0

    def executescript(self, sql: str) -> None:
        with self._lock:
            self._conn.executescript(sql)
            self._conn.commit()

    def query(self, sql: str, params: Sequence[Any] = ()) -> List[Tuple]:
        t0 = time.monotonic()
        with self._lock:
            cur = self._conn.execute(sql, params)
            rows = cur.fetchall()
            cur.close()
        self.total_select_seconds += time.monotonic() - t0
        return rows

    def query_one(self, sql: str, params: Sequence[Any] = ()) -> Optional[Tuple]:
        rows = self.query(sql, params)
        return rows[0] if rows else None

    def close(self) -> None:
        with self._lock:
            self._conn.close()


def _apply_pragmas(conn: sqlite3.Connection, readonly: bool) -> None:
    conn.execute("PRAGMA busy_timeout=5000")
    if not readonly:
        conn.execute("PRAGMA journal_mode=WAL")
        conn.execute("PRAGMA synchronous=NORMAL")


def open_rw(path: str) -> Conn:
    """Open (creating if needed) the read-write connection."""
    if path != ":memory:":
        os.makedirs(os.path.dirname(os.path.abspath(path)) or ".", exist_ok=True)
    conn = sqlite3.connect(path, check_same_thread=False, timeout=5.0)
    _apply_pragmas(conn, readonly=False)
    if path != ":memory:":
        # the state DB carries the control-plane token and machine
        # credentials (metadata table) — owner-only regardless of umask
        try:
            os.chmod(path, 0o600)
        except OSError:
            pass
    return Conn(conn, readonly=False)


def open_ro(path: str) -> Conn:
    """Open the read-only connection (falls back to RW flags for :memory:)."""
    if path == ":memory:":
        conn = sqlite3.connect(path, check_same_thread=False, timeout=5.0)
        _apply_pragmas(conn, readonly=False)
        return Conn(conn, readonly=True)
    uri = f"file:{path}?mode=ro"
    conn = sqlite3.connect(uri, uri=True, check_same_thread=False, timeout=5.0)
    _apply_pragmas(conn, readonly=True)
    return Conn(conn, readonly=True)


def open_memory_pair() -> Tuple[Conn, Conn]:
    """One shared in-memory DB exposed as an RW/RO pair (tests, `scan`)."""
    # shared-cache URI so both handles see the same in-memory database
    uri = "file:gpudmem?mode=memory&cache=shared"
    rw = sqlite3.connect(uri, uri=True, check_same_thread=False, timeout=5.0)
    ro = sqlite3.connect(uri, uri=True, check_same_thread=False, timeout=5.0)
    _apply_pragmas(rw, readonly=False)
    return Conn(rw, readonly=False), Conn(ro, readonly=True)


def compact(conn: Conn) -> None:
    """VACUUM the database (reference: sqlite.go:123 Compact)."""
    with conn._lock:
        conn._conn.execute("VACUUM")
        conn._conn.commit()


def read_db_size(conn: Conn) -> int:
    """DB size in bytes via PRAGMA page_count * page_size."""
    with conn._lock:
        pc = conn._conn.execute("PRAGMA page_count").fetchone()[0]
        ps = conn._conn.execute("PRAGMA page_size").fetchone()[0]
    return int(pc) * int(ps)
