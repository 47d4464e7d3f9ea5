"""Persistent-store (kernel crash log) scanner.

Reference: pkg/pstore/pstore.go:50-315 — scans /sys/fs/pstore files against
a match function (kernel-panic signatures), deduplicating via a SQLite
history table so each crash record is reported once; consumed by the os
component (reference: components/os/component.go:176).
"""

from __future__ import annotations

import datetime
import os
import re
from typing import Callable, List, Optional, Tuple

from .log import logger
from .sqlite_util import Conn

PSTORE_DIR = "/sys/fs/pstore"
HISTORY_TABLE = "pstore_history"

# kernel panic signatures inside pstore dmesg captures
PANIC_PATTERNS = [
    re.compile(r"Kernel panic - not syncing"),
    re.compile(r"BUG: unable to handle (?:kernel|page fault)"),
    re.compile(r"Oops: \d+"),
    re.compile(r"general protection fault"),
    re.compile(r"watchdog: BUG: soft lockup"),
    re.compile(r"NMI watchdog: Watchdog detected hard LOCKUP"),
]


def default_match(content: str) -> Optional[str]:
    for p in PANIC_PATTERNS:
        m = p.search(content)
        if m:
            return m.group(0)
    return None


class Scanner:
    def __init__(
        self,
        db_rw: Conn,
        db_ro: Conn,
        pstore_dir: str = PSTORE_DIR,
        match_fn: Callable[[str], Optional[str]] = default_match,
    ):
        self._db_rw = db_rw
        self._db_ro = db_ro
        self.pstore_dir = pstore_dir
        self._match = match_fn
        db_rw.executescript(
            f"CREATE TABLE IF NOT EXISTS {HISTORY_TABLE} ("
            "file_name TEXT NOT NULL, file_mtime INTEGER NOT NULL,"
            " matched TEXT, first_seen INTEGER NOT NULL,"
            " PRIMARY KEY (file_name, file_mtime))"
        )

    def _seen(self, name: str, mtime: int) -> bool:
        row = self._db_ro.query_one(
            f"SELECT 1 FROM {HISTORY_TABLE} WHERE file_name = ? AND file_mtime = ?",
            (name, mtime),
        )
        return row is not None

    def scan(self) -> List[Tuple[str, str, datetime.datetime]]:
        """Returns new (file_name, matched_signature, mtime) findings."""
        findings: List[Tuple[str, str, datetime.datetime]] = []
        if not os.path.isdir(self.pstore_dir):
            return findings
        try:
            names = sorted(os.listdir(self.pstore_dir))
        except OSError as e:
            logger.warning("cannot list %s: %s", self.pstore_dir, e)
            return findings
        for name in names:
            path = os.path.join(self.pstore_dir, name)
            try:
                st = os.stat(path)
                mtime = int(st.st_mtime)
                if self._seen(name, mtime):
                    continue
                with open(path, errors="replace") as f:
                    content = f.read(1024 * 1024)
            except OSError:
                continue
            matched = self._match(content)
            self._db_rw.execute(
                f"INSERT OR IGNORE INTO {HISTORY_TABLE}"
                " (file_name, file_mtime, matched, first_seen)"
                " VALUES (?, ?, ?, strftime('%s','now'))",
                (name, mtime, matched or ""),
            )
            if matched:
                findings.append(
                    (
                        name,
                        matched,
                        datetime.datetime.fromtimestamp(
                            mtime, tz=datetime.timezone.utc
                        ),
                    )
                )
        return findings
