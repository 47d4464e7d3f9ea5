"""SQLite key/value metadata store (reference: pkg/metadata/metadata.go:33-53).

Holds machine identity and credentials: machine_id, token, machine_proof,
endpoint, public/private IP, last login success timestamp.
"""

from __future__ import annotations

from typing import Dict

from .sqlite_util import Conn

TABLE = "metadata"

# Well-known keys (reference: pkg/metadata/metadata.go)
KEY_MACHINE_ID = "machine_id"
KEY_TOKEN = "token"
KEY_MACHINE_PROOF = "machine_proof"
KEY_ENDPOINT = "endpoint"
KEY_PUBLIC_IP = "public_ip"
KEY_PRIVATE_IP = "private_ip"
KEY_LOGIN_SUCCESS = "login_success_timestamp"
KEY_NODE_GROUP = "node_group"


def create_table(db_rw: Conn) -> None:
    db_rw.executescript(
        f"CREATE TABLE IF NOT EXISTS {TABLE} ("
        "key TEXT NOT NULL PRIMARY KEY, value TEXT)"
    )


def set_value(db_rw: Conn, key: str, value: str) -> None:
    db_rw.execute(
        f"INSERT INTO {TABLE} (key, value) VALUES (?, ?)"
        " ON CONFLICT(key) DO UPDATE SET value = excluded.value",
        (key, value),
    )


def get_value(db_ro: Conn, key: str) -> str:
    row = db_ro.query_one(f"SELECT value FROM {TABLE} WHERE key = ?", (key,))
    return row[0] if row and row[0] is not None else ""


def delete_value(db_rw: Conn, key: str) -> None:
    db_rw.execute(f"DELETE FROM {TABLE} WHERE key = ?", (key,))


def all_values(db_ro: Conn) -> Dict[str, str]:
    rows = db_ro.query(f"SELECT key, value FROM {TABLE} ORDER BY key")
    return {k: (v or "") for k, v in rows}
