"""HTTPS serving loop (reference: pkg/server/server.go:398-450,508,785).

Self-signed certificate generation (server.go:508 createDefaultTLSCert) via
the openssl CLI, uvicorn run in a daemon thread, and the periodic DB
compaction loop (server.go:759 doCompact).
"""

from __future__ import annotations

import os
import subprocess
import tempfile
import threading
import time
from typing import Optional, Tuple

import uvicorn

from ..bootstrap import DaemonCore
from ..pkg.log import logger
from ..pkg.sqlite_util import compact
from .app import create_app


def generate_self_signed_cert(
    directory: Optional[str] = None,
) -> Tuple[str, str]:
    """Generate an ephemeral self-signed cert; returns (cert_path, key_path)."""
    d = directory or tempfile.mkdtemp(prefix="gpud-tls-")
    os.makedirs(d, exist_ok=True)
    cert, key = os.path.join(d, "cert.pem"), os.path.join(d, "key.pem")
    if not (os.path.exists(cert) and os.path.exists(key)):
        subprocess.run(
            [
                "openssl", "req", "-x509", "-newkey", "rsa:2048",
                "-keyout", key, "-out", cert, "-days", "365", "-nodes",
                "-subj", "/CN=localhost/O=gpud",
                "-addext", "subjectAltName=DNS:localhost,IP:127.0.0.1",
            ],
            check=True,
            capture_output=True,
        )
    try:
        os.chmod(key, 0o600)  # private key owner-only regardless of umask
    except OSError:
        pass
    return cert, key


class Server:
    def __init__(
        self,
        core: DaemonCore,
        host: str = "127.0.0.1",
        port: int = 15132,
        tls: bool = True,
        plugin_specs: Optional[list] = None,
    ):
        self.core = core
        self.host = host
        self.port = port
        self.app = create_app(core, plugin_specs=plugin_specs)
        kwargs = {}
        if tls:
            cert, key = generate_self_signed_cert(
                os.path.join(core.config.data_dir, "tls")
                if os.path.isdir(core.config.data_dir)
                else None
            )
            kwargs = {"ssl_certfile": cert, "ssl_keyfile": key}
        self._uv_config = uvicorn.Config(
            self.app,
            host=host,
            port=port,
            log_level="warning",
            **kwargs,
        )
        self._uv_server = uvicorn.Server(self._uv_config)
        self._thread: Optional[threading.Thread] = None
        self._compact_stop = threading.Event()
        self._compact_thread: Optional[threading.Thread] = None

    @property
    def scheme(self) -> str:
        return "https" if self._uv_config.ssl_certfile else "http"

    @property
    def base_url(self) -> str:
        return f"{self.scheme}://{self.host}:{self.port}"

    def start(self, wait_ready: float = 10.0) -> None:
        self._thread = threading.Thread(
            target=self._uv_server.run, daemon=True, name="gpud-http"
        )
        self._thread.start()
        deadline = time.time() + wait_ready
        while time.time() < deadline and not self._uv_server.started:
            time.sleep(0.05)
        if not self._uv_server.started:
            raise RuntimeError("HTTP server failed to start")
        # periodic compaction (reference doCompact)
        period = self.core.config.compact_period_hours * 3600.0
        if period > 0:
            self._compact_thread = threading.Thread(
                target=self._compact_loop, args=(period,), daemon=True,
                name="gpud-compact",
            )
            self._compact_thread.start()

    def _compact_loop(self, period: float) -> None:
        while not self._compact_stop.wait(period):
            try:
                compact(self.core.db_rw)
                logger.info("state DB compacted")
            except Exception:
                logger.exception("DB compaction failed")

    def stop(self) -> None:
        self._compact_stop.set()
        self._uv_server.should_exit = True
        if self._thread is not None:
            self._thread.join(timeout=5.0)
