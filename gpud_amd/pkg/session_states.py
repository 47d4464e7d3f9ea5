"""Session/login state history (reference: pkg/session/states/states.go).

A small SQLite table (same name: ``session_states``) recording the last
~10 login/session outcomes so operators and the status CLI can tell "has
this node EVER failed to reach the control plane" apart from "is it fine
right now". Insert trims to the 10 most recent rows, exactly like the
reference.
"""

from __future__ import annotations

import time
from dataclasses import dataclass
from typing import List, Optional

TABLE = "session_states"
KEEP_LAST = 10


@dataclass
class State:
    timestamp: int
    success: bool
    message: str = ""


def create_table(db_rw) -> None:
    db_rw.execute(
        f"""CREATE TABLE IF NOT EXISTS {TABLE} (
            timestamp INTEGER NOT NULL,
            success INTEGER NOT NULL,
            message TEXT
        )"""
    )


def insert(db_rw, success: bool, message: str = "", timestamp: int = 0) -> None:
    ts = timestamp or int(time.time())
    db_rw.execute(
        f"INSERT INTO {TABLE} (timestamp, success, message) VALUES (?, ?, ?)",
        (ts, 1 if success else 0, message),
    )
    # retention: keep only the most recent rows (reference keeps 10)
    db_rw.execute(
        f"""DELETE FROM {TABLE} WHERE timestamp NOT IN (
            SELECT timestamp FROM {TABLE} ORDER BY timestamp DESC LIMIT ?
        )""",
        (KEEP_LAST,),
    )


def read_last(db_ro) -> Optional[State]:
    row = db_ro.query_one(
        f"SELECT timestamp, success, message FROM {TABLE} "
        f"ORDER BY timestamp DESC LIMIT 1"
    )
    if row is None:
        return None
    return State(timestamp=int(row[0]), success=bool(row[1]), message=row[2] or "")


def read_all(db_ro) -> List[State]:
    rows = db_ro.query(
        f"SELECT timestamp, success, message FROM {TABLE} ORDER BY timestamp DESC"
    )
    return [
        State(timestamp=int(r[0]), success=bool(r[1]), message=r[2] or "")
        for r in rows
    ]


def has_any_failures(db_ro) -> bool:
    row = db_ro.query_one(f"SELECT COUNT(*) FROM {TABLE} WHERE success = 0")
    return bool(row and row[0])
