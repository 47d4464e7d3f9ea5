"""Control-plane login (reference: pkg/login/login.go:29-165).

POSTs machine identity to ``{endpoint}/api/v1/login`` and persists
machine_id / token / machine_proof into the metadata table.
"""

from __future__ import annotations

import json
import uuid as uuidlib
from typing import Optional

import httpx

from . import host as pkghost
from . import metadata
from .config import Config
from .log import logger
from .sqlite_util import open_rw


def _record_state(cfg: Config, success: bool, message: str) -> None:
    """Track the outcome in the session_states table (reference:
    pkg/session/states — last 10 outcomes, queryable by gpud status)."""
    try:
        from . import session_states

        conn = open_rw(cfg.state_path)
        try:
            session_states.create_table(conn)
            session_states.insert(conn, success, message)
        finally:
            conn.close()
    except Exception:  # noqa: BLE001 — history is best-effort
        logger.debug("session state recording failed", exc_info=True)


def do_login(
    cfg: Config,
    token: str,
    endpoint: str,
    node_group: str = "",
    gpu_count: int = 0,
    timeout: float = 15.0,
    verify=None,
) -> Optional[str]:
    # login carries the bootstrap token: verified TLS by default, insecure
    # only via the explicit Config.control_plane_insecure_tls opt-in
    if verify is None:
        verify = cfg.control_plane_verify()
    machine_id = pkghost.machine_id() or str(uuidlib.uuid4())
    payload = {
        "token": token,
        "machineID": machine_id,
        "nodeGroup": node_group,
        "gpuCount": gpu_count,
        "hostname": pkghost.hostname(),
    }
    try:
        r = httpx.post(
            endpoint.rstrip("/") + "/api/v1/login",
            json=payload,
            timeout=timeout,
            verify=verify,
        )
    except httpx.HTTPError as e:
        _record_state(cfg, False, f"login request failed: {e}")
        return f"login request failed: {e}"
    if r.status_code != 200:
        msg = f"login rejected: HTTP {r.status_code} {r.text[:200]}"
        _record_state(cfg, False, msg)
        return msg
    try:
        body = r.json()
    except json.JSONDecodeError:
        body = {}
    conn = open_rw(cfg.state_path)
    try:
        metadata.create_table(conn)
        metadata.set_value(conn, metadata.KEY_MACHINE_ID, body.get("machineID", machine_id))
        metadata.set_value(conn, metadata.KEY_TOKEN, body.get("token", token))
        if body.get("machineProof"):
            metadata.set_value(conn, metadata.KEY_MACHINE_PROOF, body["machineProof"])
        metadata.set_value(conn, metadata.KEY_ENDPOINT, endpoint)
        if node_group:
            metadata.set_value(conn, metadata.KEY_NODE_GROUP, node_group)
        import time

        metadata.set_value(conn, metadata.KEY_LOGIN_SUCCESS, str(int(time.time())))
    finally:
        conn.close()
    _record_state(cfg, True, "login ok")
    logger.info("login succeeded for machine %s", machine_id)
    return None
