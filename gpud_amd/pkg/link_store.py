"""SQLite-backed link-state history with drop/flap detection.

Reference: components/accelerator/nvidia/infiniband/store (store.go:57-319,
scan_drops.go, scan_flaps.go, events.go) — a per-(device, port) time series
of link states persisted in SQLite so drop/flap evaluation SURVIVES daemon
restarts, with:

  * rate-limited inserts (min 15 s between snapshots);
  * **drop** detection: a port continuously down for >= the drop threshold
    (default 4 min);
  * **flap** detection: a port that stays down for >= the flap
    down-interval (default 25 s) and then reverts to active, at least
    flap-back-to-active-threshold times (default 3) within the scan window;
  * event marking on the history rows (event_type/event_reason), a
    tombstone set by ``set-healthy`` so cleared events stay cleared across
    restarts, and retention purging (default 5 days).

Used by both the xGMI component (the MI355X fabric — links are "ports" of
a GPU "device") and the infiniband component (IB ports per HCA).
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import Callable, Dict, List, Optional, Tuple

DEFAULT_MIN_INSERT_INTERVAL = 15.0  # reference: insert.go:17
DEFAULT_RETENTION_SECONDS = 5 * 24 * 3600.0  # reference: purge.go:15
DEFAULT_DROP_THRESHOLD = 4 * 60.0  # reference: scan_drops.go:14
DEFAULT_FLAP_DOWN_INTERVAL = 25.0  # reference: scan_flaps.go:13
DEFAULT_FLAP_BACK_THRESHOLD = 3  # reference: scan_flaps.go:17

EVENT_DROP = "link_port_drop"  # ≙ reference EventTypeIbPortDrop
EVENT_FLAP = "link_port_flap"  # ≙ reference EventTypeIbPortFlap

STATE_ACTIVE = "active"
STATE_DOWN = "down"


@dataclass
class LinkEvent:
    ts: float
    device: str
    port: int
    event_type: str
    reason: str


class LinkStore:
    def __init__(
        self,
        db_rw,
        db_ro,
        table_prefix: str = "xgmi",
        get_time_now: Callable[[], float] = time.time,
        min_insert_interval: float = DEFAULT_MIN_INSERT_INTERVAL,
        retention_seconds: float = DEFAULT_RETENTION_SECONDS,
        drop_threshold: float = DEFAULT_DROP_THRESHOLD,
        flap_down_interval: float = DEFAULT_FLAP_DOWN_INTERVAL,
        flap_back_threshold: int = DEFAULT_FLAP_BACK_THRESHOLD,
    ):
        self.db_rw = db_rw
        self.db_ro = db_ro
        self.table = f"{table_prefix}_link_history_v0_1"
        self.meta_table = f"{table_prefix}_link_metadata_v0_1"
        self.now = get_time_now
        self.min_insert_interval = min_insert_interval
        self.retention_seconds = retention_seconds
        self.drop_threshold = drop_threshold
        self.flap_down_interval = flap_down_interval
        self.flap_back_threshold = flap_back_threshold
        self._last_insert_ts = 0.0
        self.db_rw.execute(
            f"""CREATE TABLE IF NOT EXISTS {self.table} (
                    ts INTEGER NOT NULL,
                    device TEXT NOT NULL,
                    port INTEGER NOT NULL,
                    state TEXT NOT NULL,
                    rate_gb_sec REAL DEFAULT 0,
                    total_link_downed INTEGER DEFAULT 0,
                    event_type TEXT DEFAULT '',
                    event_reason TEXT DEFAULT ''
                )"""
        )
        self.db_rw.execute(
            f"CREATE INDEX IF NOT EXISTS idx_{self.table}_dev_port_ts "
            f"ON {self.table} (device, port, ts)"
        )
        self.db_rw.execute(
            f"""CREATE TABLE IF NOT EXISTS {self.meta_table} (
                key TEXT PRIMARY KEY, value TEXT
            )"""
        )

    # -- inserts --------------------------------------------------------------

    def insert(self, snapshots: List[Dict]) -> bool:
        """Insert one sweep of {device, port, state[, rate_gb_sec,
        total_link_downed]} rows; rate-limited (reference: minInsertInterval
        prevents excessive inserts). Returns whether rows were written."""
        now = self.now()
        if now - self._last_insert_ts < self.min_insert_interval:
            return False
        self._last_insert_ts = now
        self.db_rw.executemany(
                f"INSERT INTO {self.table} (ts, device, port, state, "
                "rate_gb_sec, total_link_downed) VALUES (?, ?, ?, ?, ?, ?)",
            [
                (
                    int(now),
                    s["device"],
                    int(s["port"]),
                    s["state"],
                    float(s.get("rate_gb_sec", 0.0)),
                    int(s.get("total_link_downed", 0)),
                )
                for s in snapshots
            ],
        )
        return True

    # -- tombstone (set-healthy survives restarts) ----------------------------

    def set_tombstone(self, ts: Optional[float] = None) -> None:
        ts = self.now() if ts is None else ts
        self.db_rw.execute(
            f"INSERT INTO {self.meta_table} (key, value) VALUES "
            "('tombstone_ts', ?) ON CONFLICT(key) DO UPDATE SET value "
            "= excluded.value",
            (str(int(ts)),),
        )

    def get_tombstone(self) -> float:
        row = self.db_ro.query_one(
            f"SELECT value FROM {self.meta_table} WHERE key = 'tombstone_ts'"
        )
        return float(row[0]) if row else 0.0

    # -- scans ----------------------------------------------------------------

    def _snapshots(self, since: float) -> Dict[Tuple[str, int], List[Tuple[float, str]]]:
        out: Dict[Tuple[str, int], List[Tuple[float, str]]] = {}
        for ts, dev, port, state in self.db_ro.query(
            f"SELECT ts, device, port, state FROM {self.table} "
            "WHERE ts > ? ORDER BY ts ASC",
            (int(since),),
        ):
            out.setdefault((dev, port), []).append((float(ts), state))
        return out

    def _mark(self, dev: str, port: int, ts: float, event_type: str,
              reason: str) -> None:
        self.db_rw.execute(
            f"UPDATE {self.table} SET event_type = ?, event_reason = ? "
            "WHERE device = ? AND port = ? AND ts = ?",
            (event_type, reason, dev, port, int(ts)),
        )

    @staticmethod
    def _find_drops(
        series: List[Tuple[float, str]], drop_threshold: float
    ) -> List[Tuple[float, float]]:
        """(down_since, last_down_ts) for runs of consecutive 'down'
        snapshots spanning >= drop_threshold (reference: scan_drops.go)."""
        drops = []
        run_start = None
        last_down = None
        for ts, state in series:
            if state == STATE_DOWN:
                if run_start is None:
                    run_start = ts
                last_down = ts
            else:
                if run_start is not None and last_down is not None:
                    if last_down - run_start >= drop_threshold:
                        drops.append((run_start, last_down))
                run_start = None
                last_down = None
        if run_start is not None and last_down is not None:
            if last_down - run_start >= drop_threshold:
                drops.append((run_start, last_down))
        return drops

    @staticmethod
    def _find_flaps(
        series: List[Tuple[float, str]],
        down_interval: float,
        back_threshold: int,
    ) -> List[Tuple[float, float]]:
        """(down_since, reverted_at) for active-reverts after persistent
        down runs; only returned when there are >= back_threshold of them
        (reference: scan_flaps.go findFlaps — down for more than the
        interval, flapped back to active, more than N times)."""
        if len(series) < 3 or len(series) < back_threshold:
            return []
        reverts = []
        run_start = None
        last_down = None
        for ts, state in series:
            if state == STATE_DOWN:
                if run_start is None:
                    run_start = ts
                last_down = ts
            elif state == STATE_ACTIVE:
                if (
                    run_start is not None
                    and last_down is not None
                    and last_down - run_start >= down_interval
                ):
                    reverts.append((run_start, ts))
                run_start = None
                last_down = None
        return reverts if len(reverts) >= back_threshold else []

    def scan(self) -> List[LinkEvent]:
        """Evaluate the whole window (retention, bounded below by the
        tombstone), mark event rows, and return the events found."""
        now = self.now()
        since = max(now - self.retention_seconds, self.get_tombstone())
        events: List[LinkEvent] = []
        for (dev, port), series in self._snapshots(since).items():
            for down_since, last_down in self._find_drops(
                series, self.drop_threshold
            ):
                reason = (
                    f"{dev} port {port} down since "
                    f"{time.strftime('%Y-%m-%dT%H:%M:%SZ', time.gmtime(down_since))}"
                )
                self._mark(dev, port, last_down, EVENT_DROP, reason)
                events.append(LinkEvent(last_down, dev, port, EVENT_DROP, reason))
            for down_since, reverted_at in self._find_flaps(
                series, self.flap_down_interval, self.flap_back_threshold
            ):
                reason = (
                    f"{dev} port {port} down since "
                    f"{time.strftime('%Y-%m-%dT%H:%M:%SZ', time.gmtime(down_since))}"
                    " (and flapped back to active)"
                )
                self._mark(dev, port, reverted_at, EVENT_FLAP, reason)
                events.append(
                    LinkEvent(reverted_at, dev, port, EVENT_FLAP, reason)
                )
        return events

    def latest_state(self, dev: str, port: int) -> Optional[str]:
        row = self.db_ro.query_one(
            f"SELECT state FROM {self.table} WHERE device = ? AND port = ? "
            "ORDER BY ts DESC LIMIT 1",
            (dev, port),
        )
        return row[0] if row else None

    def evaluate(
        self,
        drop_sticky_window: float = 10 * 60.0,
        flap_auto_clear_window: float = 0.0,
    ) -> Dict[str, List[LinkEvent]]:
        """Health view over the scanned events (reference: the component's
        drop sticky window — unhealthy for a stabilization period after
        recovery, default 10 min — and the flap auto-clear window, default
        0 = sticky until set-healthy)."""
        now = self.now()
        events = self.scan()
        active_drops: List[LinkEvent] = []
        active_flaps: List[LinkEvent] = []
        seen = set()
        # newest first so the per-(device,port) representative event is the
        # most recent one (the auto-clear windows key off it)
        events.sort(key=lambda e: e.ts, reverse=True)
        for ev in events:
            key = (ev.device, ev.port, ev.event_type)
            if key in seen:
                continue
            if ev.event_type == EVENT_DROP:
                still_down = self.latest_state(ev.device, ev.port) == STATE_DOWN
                if still_down or now - ev.ts <= drop_sticky_window:
                    seen.add(key)
                    active_drops.append(ev)
            elif ev.event_type == EVENT_FLAP:
                if flap_auto_clear_window <= 0 or now - ev.ts <= flap_auto_clear_window:
                    seen.add(key)
                    active_flaps.append(ev)
        return {"drops": active_drops, "flaps": active_flaps}

    # -- retention ------------------------------------------------------------

    def purge(self) -> int:
        cutoff = self.now() - self.retention_seconds
        row = self.db_ro.query_one(
            f"SELECT COUNT(*) FROM {self.table} WHERE ts < ?", (int(cutoff),)
        )
        n = int(row[0]) if row else 0
        self.db_rw.execute(
            f"DELETE FROM {self.table} WHERE ts < ?", (int(cutoff),)
        )
        return n
